import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from metaflow_amd.ops import kernels as K
ext = K.hip_ext()
torch.manual_seed(0)
# asymmetric random inputs (guide G9: transpose-detecting)
a = torch.randn(32, 16, dtype=torch.bfloat16, device="cuda")
b = torch.randn(16, 32, dtype=torch.bfloat16, device="cuda")
c = ext.dbg_mfma32(a, b)
ref = a.float() @ b.float()
err = ((c - ref).norm() / ref.norm()).item()
errT = ((c - ref.T).norm() / ref.norm()).item()
print("mfma32 layout: err=%.2e  errT=%.2e  -> %s" % (err, errT,
      "OK" if err < 1e-2 else ("TRANSPOSED" if errT < 1e-2 else "WRONG")))
