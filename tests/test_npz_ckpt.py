"""T2: .npz checkpoint format (C2): numpy interop,
byte-stability, save->load->resume determinism."""
import hashlib
import zipfile

import numpy as np
import pytest

import nezha_amd as nz


def test_npz_numpy_interop(cpu_ctx, tmp_path):
    tr = nz.Trainer.mlp(cpu_ctx, layers=2, dim=32, batch=8, seed=11)
    p = str(tmp_path / "ck.npz")
    tr.save(p)
    ck = np.load(p)
    names = set(ck.files)
    assert "param/mlp.l0.w" in names
    assert "master/mlp.l0.w" in names
    assert "adam_m/mlp.l1.b" in names
    assert "adam_v/mlp.l1.b" in names
    assert "meta/step" in names
    assert ck["param/mlp.l0.w"].shape == (32, 32)
    assert ck["param/mlp.l0.w"].dtype == np.float32  # CPU params are f32
    # a stored (uncompressed) real zip
    with zipfile.ZipFile(p) as z:
        for info in z.infolist():
            assert info.compress_type == zipfile.ZIP_STORED


def test_npz_byte_stable(cpu_ctx, tmp_path):
    tr = nz.Trainer.mlp(cpu_ctx, layers=2, dim=32, batch=8, seed=11)
    p1, p2 = str(tmp_path / "a.npz"), str(tmp_path / "b.npz")
    tr.save(p1)
    tr.save(p2)
    h1 = hashlib.sha256(open(p1, "rb").read()).hexdigest()
    h2 = hashlib.sha256(open(p2, "rb").read()).hexdigest()
    assert h1 == h2


def test_npz_roundtrip_via_numpy_write(cpu_ctx, tmp_path):
    """We can load an npz that numpy wrote (vice-versa interop)."""
    tr = nz.Trainer.mlp(cpu_ctx, layers=2, dim=16, batch=4, seed=2)
    p = str(tmp_path / "ours.npz")
    tr.save(p)
    ck = dict(np.load(p))
    # perturb a param, write with numpy (uncompressed), load into the engine
    ck["param/mlp.l0.w"] = ck["param/mlp.l0.w"] * 0 + 0.5
    ck["master/mlp.l0.w"] = ck["master/mlp.l0.w"] * 0 + 0.5
    p2 = str(tmp_path / "numpy.npz")
    np.savez(p2, **ck)
    tr.load(p2)
    p3 = str(tmp_path / "resaved.npz")
    tr.save(p3)
    ck3 = np.load(p3)
    assert np.all(ck3["param/mlp.l0.w"] == 0.5)


def test_ckpt_resume_determinism(cpu_ctx, tmp_path):
    """Train 3 steps, snapshot, train 3 more; resume from snapshot in a fresh
    trainer and verify bitwise-identical parameters (CPU path)."""
    kw = dict(layers=2, dim=32, batch=8, lr=1e-3, seed=21)
    a = nz.Trainer.mlp(cpu_ctx, **kw)
    for _ in range(3):
        a.step()
    snap = str(tmp_path / "snap.npz")
    a.save(snap)
    for _ in range(3):
        a.step()
    final_a = str(tmp_path / "fa.npz")
    a.save(final_a)

    b = nz.Trainer.mlp(cpu_ctx, **kw)
    b.load(snap)
    assert b.step_count == 3
    for _ in range(3):
        b.step()
    final_b = str(tmp_path / "fb.npz")
    b.save(final_b)

    ca, cb = np.load(final_a), np.load(final_b)
    for k in ca.files:
        np.testing.assert_array_equal(ca[k], cb[k], err_msg=k)


def test_load_shape_mismatch_rejected(tmp_path):
    """Loading a checkpoint into a differently-shaped model must raise a
    clean error, not corrupt memory."""
    ctx = nz.Ctx(nz.CPU)
    a = nz.Trainer.mlp(ctx, layers=2, dim=64, batch=16, lr=1e-3, seed=1)
    a.step()
    p = str(tmp_path / "a.npz")
    a.save(p)
    b = nz.Trainer.mlp(ctx, layers=2, dim=128, batch=16, lr=1e-3, seed=1)
    with pytest.raises(nz.NezhaError, match="mismatch|missing"):
        b.load(p)


def test_corrupt_npz_rejected(tmp_path):
    """Truncated / bit-flipped checkpoints must raise clean errors."""
    ctx = nz.Ctx(nz.CPU)
    a = nz.Trainer.mlp(ctx, layers=2, dim=64, batch=16, lr=1e-3, seed=1)
    a.step()
    p = tmp_path / "a.npz"
    a.save(str(p))
    raw = p.read_bytes()
    # truncation
    (tmp_path / "t.npz").write_bytes(raw[: len(raw) // 2])
    b = nz.Trainer.mlp(ctx, layers=2, dim=64, batch=16, lr=1e-3, seed=1)
    with pytest.raises(nz.NezhaError):
        b.load(str(tmp_path / "t.npz"))
    # bit flip in the middle (crc or structure must catch it)
    flipped = bytearray(raw)
    flipped[len(raw) // 3] ^= 0xFF
    (tmp_path / "f.npz").write_bytes(bytes(flipped))
    with pytest.raises(nz.NezhaError):
        b.load(str(tmp_path / "f.npz"))
    # b must still be usable after failed loads
    b.step()
