"""Cache-aliasing regression: torch's caching allocator
reuses freed addresses, so a cast cache keyed by raw data_ptr could serve a
stale f32 copy after a model is freed and a same-shape tensor lands at the
recycled address.  The fix pins the source tensor inside the cache entry so
its address stays out of the free pool while the entry lives — these tests
prove both the pinning and the absence of stale hits under aggressive
free/realloc churn (CPU tensors; the cache code is device-agnostic)."""

import pytest

torch = pytest.importorskip("torch")

gpu_ops = pytest.importorskip("traffic_classifier_sdn_amd.ops.gpu")


def setup_function(_):
    gpu_ops._cast_cache.clear()
    gpu_ops._gnb_cache.clear()
    gpu_ops._knn_cmean_cache.clear()


def test_cast_cache_pins_source_buffer():
    t = torch.arange(24, dtype=torch.float64).reshape(2, 12)
    c = gpu_ops._f32_cached(t)
    assert torch.equal(c, t.float())
    key = (t.data_ptr(), t.numel(), t.dtype)
    src, cached = gpu_ops._cast_cache[key]
    assert src is t  # the entry holds the source => address cannot be recycled
    del t
    # entry keeps the buffer alive; contents still the original model's
    assert torch.equal(gpu_ops._cast_cache[key][1], cached)


def test_no_stale_cast_after_free_and_realloc_churn():
    # free a "model", then allocate many same-shape tensors: none may be
    # served another tensor's cached cast
    for i in range(50):
        t = torch.full((6, 12), float(i), dtype=torch.float64)
        c = gpu_ops._f32_cached(t)
        assert torch.equal(c, torch.full((6, 12), float(i), dtype=torch.float32)), i
        del t, c


def test_gnb_cache_entry_pins_sources(monkeypatch):
    # the HIP gnb_predict kernel needs device tensors, so stub the extension
    # call and drive the REAL gnb_argmax caching logic on CPU tensors: the
    # entry it builds must hold the var/prior source tensors (pinning their
    # addresses against recycling)
    captured = {}

    def fake_predict(X, theta32, inv_var, const32):
        captured["inv_var"] = inv_var
        return torch.zeros(X.shape[0], dtype=torch.int64)

    monkeypatch.setattr(gpu_ops._ext, "gnb_predict", fake_predict)
    X = torch.randn(8, 12)
    theta = torch.randn(6, 12)
    var = torch.rand(6, 12) + 0.5
    prior = torch.full((6,), 1 / 6.0)
    gpu_ops.gnb_argmax(X, theta, var, prior)
    ent = gpu_ops._gnb_cache[(var.data_ptr(), prior.data_ptr())]
    assert ent[0] is var and ent[1] is prior
    assert torch.allclose(captured["inv_var"].double(), 1.0 / var.double(), rtol=1e-6)
    # second call with the same model hits the cache (same derived tensors)
    gpu_ops.gnb_argmax(X, theta, var, prior)
    assert captured["inv_var"] is ent[3]


def test_knn_cmean_no_stale_after_refit_churn():
    for i in range(1, 6):
        R = torch.full((100, 12), float(i), dtype=torch.float32)
        cm = gpu_ops._knn_cmean(R)
        assert torch.allclose(cm, torch.full((12,), float(i))), i
        del R
