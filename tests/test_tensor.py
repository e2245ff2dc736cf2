"""T1: tensor engine basics on CPU."""
import numpy as np

import nezha_amd as nz


def test_create_fill_download(cpu_ctx):
    t = cpu_ctx.tensor((4, 8), nz.F32)
    t.fill(3.5)
    a = t.numpy()
    assert a.shape == (4, 8)
    assert np.all(a == 3.5)


def test_upload_roundtrip(cpu_ctx):
    x = np.random.default_rng(0).normal(size=(16, 16)).astype(np.float32)
    t = cpu_ctx.from_numpy(x)
    np.testing.assert_array_equal(t.numpy(), x)


def test_bf16_conversion_roundtrip(cpu_ctx):
    x = np.array([1.0, -2.5, 3.141592, 1e-8, 65504.0, -1e30], dtype=np.float32)
    bits = nz.f32_to_bf16(x)
    back = nz.bf16_to_f32(bits)
    assert np.all(np.abs(back - x) <= np.abs(x) * 0.01)


def test_bf16_tensor(cpu_ctx):
    x = np.random.default_rng(1).normal(size=(8, 8)).astype(np.float32)
    t = cpu_ctx.from_numpy_bf16(x)
    y = t.numpy()
    assert np.allclose(y, x, atol=0.02, rtol=0.01)


def test_rng_determinism(cpu_ctx):
    a = cpu_ctx.tensor((1000,), nz.F32).rand_normal(42).numpy()
    b = cpu_ctx.tensor((1000,), nz.F32).rand_normal(42).numpy()
    c = cpu_ctx.tensor((1000,), nz.F32).rand_normal(43).numpy()
    np.testing.assert_array_equal(a, b)
    assert not np.array_equal(a, c)
    # sanity: roughly standard normal
    assert abs(a.mean()) < 0.15 and abs(a.std() - 1.0) < 0.15


def test_rand_uniform_range(cpu_ctx):
    a = cpu_ctx.tensor((5000,), nz.F32).rand_uniform(7, -2.0, 3.0).numpy()
    assert a.min() >= -2.0 and a.max() < 3.0
    assert abs(a.mean() - 0.5) < 0.2


def test_bench_script_parses():
    """bench.py is the driver contract — it must always at least parse."""
    import ast, os
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    for f in ("bench.py", "__graft_entry__.py"):
        ast.parse(open(os.path.join(repo, f)).read())
