"""Greedy speculative decoding: a small DRAFT model proposes k tokens,
the TARGET model verifies them in one batched forward.

Token-exact with plain target-greedy by construction (greedy
acceptance: keep the longest proposal prefix the target would itself
have chosen, then emit the target's own next token — per step at least
one target token of progress, at most k+1). No reference counterpart
(serving is a new capability, SURVEY §2.4 note); the MI355X angle is
that verification replaces k launch-bound single-token decode steps
with ONE k+1-token forward through the flash/cache attention path, so
decode becomes compute-shaped instead of latency-shaped.

Cache bookkeeping leans on KVCache.pos being the single read boundary
(models/llama.py): rejected proposal rows stay in the cache allocation
but are never read once ``pos`` is rolled back, so rollback is free.
"""

def _greedy_row(logits):
    # logits [1, S, V] -> [S] int64 greedy tokens
    return logits[0].float().argmax(dim=-1)


def speculative_generate(target, draft, tokens, max_new_tokens,
                         k=4):
    """Greedy decode ``max_new_tokens`` continuation tokens for ONE
    sequence (tokens [1, S0]) using draft-proposal/target-verify.

    Returns (out_tokens [1, S0+max_new], stats) where stats is a dict
    with ``proposed``/``accepted``/``target_steps`` for measuring the
    acceptance rate. Output is identical to ``target.generate(tokens,
    max_new_tokens)`` (greedy) — asserted by tests, guaranteed by the
    acceptance rule."""
    # torch imported lazily: this module is exported from the package
    # root, and a module-level torch import would load torch into every
    # small task subprocess (the bisected 10x startup regression)
    import torch

    from .models.llama import KVCache

    assert tokens.dim() == 2 and tokens.size(0) == 1, \
        "speculative_generate is per-sequence (B=1)"
    with torch.no_grad():
        return _speculative_generate(torch, target, draft, tokens,
                                     max_new_tokens, k)


def _speculative_generate(torch, target, draft, tokens,
                          max_new_tokens, k):
    from .models.llama import KVCache
    device = tokens.device
    S0 = tokens.size(1)
    cap = S0 + max_new_tokens + k + 2
    t_cache = KVCache(target.cfg, 1, cap, device,
                      dtype=target.embed.weight.dtype)
    d_cache = KVCache(draft.cfg, 1, cap, device,
                      dtype=draft.embed.weight.dtype)

    # prefill both models; the target's last-position logits give the
    # first emitted token (same as generate()'s first step)
    t_logits = target(tokens, cache=t_cache)
    draft(tokens, cache=d_cache)
    out = [int(_greedy_row(t_logits)[-1])]
    stats = {"proposed": 0, "accepted": 0, "target_steps": 1}

    # tokens the draft has NOT consumed yet (the freshly emitted one)
    pending_draft = [out[-1]]

    while len(out) < max_new_tokens:
        budget = max_new_tokens - len(out)
        kk = min(k, budget)
        # ---- draft proposes kk tokens autoregressively
        proposal = []
        feed = pending_draft
        for _ in range(kk):
            d_logits = draft(torch.tensor([feed], device=device),
                             cache=d_cache)
            nxt = int(_greedy_row(d_logits)[-1])
            proposal.append(nxt)
            feed = [nxt]
        stats["proposed"] += kk

        # ---- target verifies [emitted_last] + proposal in ONE forward:
        # position i's logits give the target's choice AFTER seeing
        # proposal[:i]
        block = [out[-1]] + proposal
        t_pos_before = t_cache.pos
        t_logits = target(torch.tensor([block], device=device),
                          cache=t_cache)
        stats["target_steps"] += 1
        choice = _greedy_row(t_logits)  # [kk+1]

        accepted = 0
        while accepted < kk and int(choice[accepted]) == \
                proposal[accepted]:
            accepted += 1
        emitted = proposal[:accepted] + [int(choice[accepted])]
        out.extend(emitted)
        stats["accepted"] += accepted

        # ---- roll caches back to exactly the emitted history.
        # target consumed block (1+kk rows); keep 1+accepted of them
        # (out[-1] row + accepted proposal rows); choice[accepted] is
        # NOT cached yet — it is the next round's first block token.
        t_cache.pos = t_pos_before + 1 + accepted
        # draft consumed out[-1] + proposal[:-1]; keep rows for tokens
        # up to the last ACCEPTED proposal and recompute the rest next
        # round from the pending (unconsumed) history suffix
        d_keep = t_pos_before + 1 + accepted
        if d_cache.pos > d_keep:
            # some proposals rejected: discard their rows; the draft
            # has consumed everything accepted, only the new target
            # emission is pending
            d_cache.pos = d_keep
            pending_draft = [emitted[-1]]
        else:
            # all kk proposals accepted: the draft never consumed its
            # own LAST proposal (it only emitted it), so both that
            # token and the target's bonus emission are pending
            pending_draft = emitted[-2:] if accepted == kk and kk > 0 \
                else [emitted[-1]]

    out = out[:max_new_tokens]
    return torch.cat(
        [tokens, torch.tensor([out], device=device)], dim=1), stats
