"""HBM-capacity demo: 1B-row (48 GB) KNN reference shard resident on ONE
MI355X; brute-force top-k over it with the MFMA kernel."""
import sys, time
sys.path.insert(0, ".")
import torch
from traffic_classifier_sdn_amd.ops import gpu as og

n_ref, n_q, k = 1_000_000_000, 16_384, 5
g = torch.Generator(device="cuda").manual_seed(0)
R = torch.rand(n_ref, 12, device="cuda", generator=g) * 1e5
Q = torch.rand(n_q, 12, device="cuda", generator=g) * 1e5
torch.cuda.synchronize()
print(f"resident reference: {R.numel()*4/2**30:.1f} GiB on", torch.cuda.get_device_name(0))
res = {}
for approx in (False, True):
    d, i = og.knn_topk(Q, R, k, approx=approx)  # warmup
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    d, i = og.knn_topk(Q, R, k, approx=approx)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    tag = "bf16-coarse+exact-refine" if approx else "exact-f32"
    print(f"[{tag}] top-{k} of {n_q} queries vs {n_ref/1e9:.0f}B rows: "
          f"{dt:.2f}s = {n_q*n_ref/dt:.3g} candidate distances/s")
    assert int(i.max()) < n_ref and int(i.min()) >= 0
    res[approx] = i.cpu()
# recall of the coarse pass at the full 1B-row scale
ex, ap = res[False].numpy(), res[True].numpy()
hits = sum(len(set(ex[q]).intersection(ap[q])) for q in range(n_q))
print(f"bf16-coarse recall@{k} vs exact over 1B rows: {hits/(n_q*k):.6f}")
print("OK")
