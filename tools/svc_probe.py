import sys
sys.path.insert(0, ".")
import numpy as np, torch
from traffic_classifier_sdn_amd.models import SVC
from traffic_classifier_sdn_amd.ops import cpu as oc
from bench import _stratified_synth_rows

Xn, lab, classes = _stratified_synth_rows(1_000_000, 0)
yn = classes[lab]
m = SVC(tol=1e-3, max_iter=60_000, device="cuda").fit(Xn, yn)
print("n_sv", int(m.n_support_.sum()))
for ne in (4_000, 50_000):
    Xe, lab_e, _ = _stratified_synth_rows(ne, 31337)
    ye = classes[lab_e].astype(str)
    pred_gpu = m.predict(Xe).astype(str)
    sub = np.random.default_rng(0).choice(ne, 3000, replace=False)
    Xt = torch.from_numpy(Xe[sub]).float()
    dec = oc.svc_predict(Xt, m.support_vectors_.float().cpu(), m.dual_coef_.cpu(),
                         m.intercept_.cpu(), m.n_support_.cpu(), m.gamma_)
    pred_cpu = m.classes_[dec.numpy()].astype(str)
    print(f"ne={ne}: acc_gpu={(pred_gpu==ye).mean():.4f} "
          f"acc_cpu(sub)={(pred_cpu==ye[sub]).mean():.4f} agree(sub)={(pred_gpu[sub]==pred_cpu).mean():.4f}")
