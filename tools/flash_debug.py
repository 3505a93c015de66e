"""Diagnose flash_prefill failures: infer the kernel's effective causal
mask (K=0, one-hot V => output row = histogram of attended kv tokens),
then per-prior value comparison. Run on a GPU box."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(
    os.path.abspath(__file__)), ".."))

import torch  # noqa: E402

from llm_d_inference_scheduler_amd.ops import hip_ops

ext = hip_ops()
D, BS = 128, 16


def run_case(chunk, prior, qpg, kvh, probe_mask=False, seed=0):
    torch.manual_seed(seed)
    ctx = prior + chunk
    qh = kvh * qpg
    nb = (ctx + BS - 1) // BS
    q = torch.randn(chunk, qh, D, device="cuda").bfloat16()
    kc = torch.randn(nb + 1, kvh, BS, D, device="cuda").bfloat16()
    vc = torch.randn(nb + 1, kvh, BS, D, device="cuda").bfloat16()
    if probe_mask:
        kc.zero_()
        vc.zero_()
        for t in range(ctx):
            vc[1 + t // BS, :, t % BS, t % D] = 1.0
    bt = torch.arange(1, nb + 1, dtype=torch.int32, device="cuda").view(1, -1)
    meta = torch.tensor([[0, chunk, prior]], dtype=torch.int32, device="cuda")
    tiles = torch.tensor([(0, v) for v in range(0, chunk * qpg, 128)],
                         dtype=torch.int32, device="cuda")
    out = ext.flash_prefill(q, kc, vc, bt, meta, tiles, D ** -0.5)

    if probe_mask:
        # expected: out[p][h][d] = count(valid kv == d mod 128)/count
        bad = 0
        for p in range(chunk):
            n_valid = prior + p + 1
            expect = torch.zeros(D)
            for t in range(n_valid):
                expect[t % D] += 1.0 / n_valid
            got = out[p, 0].float().cpu()
            if (got - expect).abs().max() > 0.02:
                if bad < 4:
                    err_d = int((got - expect).abs().argmax())
                    print(f"  p={p} (n_valid={n_valid}): maxerr "
                          f"{(got-expect).abs().max():.3f} at dim {err_d} "
                          f"got {got[err_d]:.3f} want {expect[err_d]:.3f}")
                bad += 1
        print(f"mask probe chunk={chunk} prior={prior} qpg={qpg}: "
              f"{bad}/{chunk} rows wrong")
        return bad == 0

    # full value comparison vs fp32 reference
    kk = kc[1:].float().permute(1, 0, 2, 3).reshape(kvh, -1, D)[:, :ctx]
    vv = vc[1:].float().permute(1, 0, 2, 3).reshape(kvh, -1, D)[:, :ctx]
    qi = q.float().view(chunk, kvh, qpg, D).permute(1, 2, 0, 3)
    s = torch.einsum("hgtd,hsd->hgts", qi, kk) * D ** -0.5
    t_idx = torch.arange(chunk, device="cuda").view(1, 1, -1, 1)
    s_idx = torch.arange(ctx, device="cuda").view(1, 1, 1, -1)
    s.masked_fill_(s_idx > t_idx + prior, float("-inf"))
    o = torch.einsum("hgts,hsd->hgtd", torch.softmax(s, -1), vv)
    ref = o.permute(2, 0, 1, 3).reshape(chunk, qh, D)
    d = (out.float() - ref).abs()
    per_row = d.amax(dim=(1, 2)).cpu()
    worst = per_row.argmax()
    print(f"values chunk={chunk} prior={prior} qpg={qpg} kvh={kvh}: "
          f"max={d.max():.4f} worst_row={int(worst)} "
          f"rows>{0.03}: {[int(i) for i in (per_row > 0.03).nonzero()[:12]]}")
    return float(d.max()) < 0.03


if __name__ == "__main__":
    torch.cuda.init()
    ok = True
    for prior in [0, 5, 16, 17, 32, 100, 171, 512]:
        ok &= run_case(64, prior, 4, 2, probe_mask=True)
    for prior in [0, 16, 171, 512]:
        for qpg in [1, 4, 8]:
            ok &= run_case(64, prior, qpg, 2)
    ok &= run_case(257, 512, 8, 8)
    ok &= run_case(333, 171, 4, 8)
    print("ALL OK" if ok else "FAILURES ABOVE")
    sys.exit(0 if ok else 1)
