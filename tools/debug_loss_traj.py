import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from metaflow_amd.models.llama import LlamaConfig, LlamaForCausalLM
from metaflow_amd.parallel.ddp import FlatParamModel, FusedAdamW

for batch in (4, 8):
    torch.manual_seed(1234)
    with torch.device("cuda:0"):
        model = LlamaForCausalLM(LlamaConfig.llama3_8b())
    flat = FlatParamModel(model)
    opt = FusedAdamW(flat, lr=3e-4)
    torch.manual_seed(5678)
    tok = torch.randint(0, 128256, (batch, 4097), device="cuda:0")
    losses = []
    for i in range(6):
        flat.zero_grad()
        loss = model(tok[:, :-1], tok[:, 1:].contiguous())
        loss.backward()
        flat.finish_grad_sync()
        opt.step()
        losses.append(round(float(loss.item()), 3))
    print("batch", batch, "losses", losses)
    del model, flat, opt
    torch.cuda.empty_cache()
