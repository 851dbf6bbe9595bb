"""TF bundle checkpoint reader vs the shipped reference model (copied into
artifacts/reference_ckpt — 3,361 fp64 params, K=1)."""
import os

import numpy as np
import pytest
import torch

PREFIX = os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "artifacts", "reference_ckpt",
    "cp-0000.ckpt")

pytestmark = pytest.mark.skipif(not os.path.isfile(PREFIX + ".index"),
                                reason="reference checkpoint not present")


def test_read_bundle_structure():
    from multihop_offload_amd.utils.tfckpt import read_bundle
    t = read_bundle(PREFIX)
    weights = {k: v for k, v in t.items() if "VARIABLE_VALUE" in k}
    assert len(weights) == 10                       # 5 layers × {kernel,bias}
    assert sum(v.size for v in weights.values()) == 3361
    k0 = weights["layer_with_weights-0/kernel/.ATTRIBUTES/VARIABLE_VALUE"]
    assert k0.shape == (1, 4, 32) and k0.dtype == np.float64
    k4 = weights["layer_with_weights-4/kernel/.ATTRIBUTES/VARIABLE_VALUE"]
    assert k4.shape == (1, 32, 1)
    # all offsets/sizes must tile the data shard exactly
    data_len = os.path.getsize(PREFIX + ".data-00000-of-00001")
    assert sum(v.nbytes for v in weights.values()) <= data_len


def test_load_into_model_and_forward(small_case, jobs_for):
    from multihop_offload_amd.models.chebconv import ChebConvStack
    from multihop_offload_amd.utils.tfckpt import load_reference_weights
    from multihop_offload_amd.queueing import ConflictCSR

    m = ChebConvStack(K=1, dtype=torch.float64)
    load_reference_weights(m, PREFIX)
    g = small_case
    x = torch.tensor(g.ext.features(jobs_for))
    sup = ConflictCSR(g.ext.ext_indptr, g.ext.ext_indices)
    lam = m(x, sup)
    assert lam.shape == (g.ext.num_edges_ext, 1)
    assert torch.isfinite(lam).all() and (lam >= 0).all()
    # K mismatch must be detected
    m2 = ChebConvStack(K=2, dtype=torch.float64)
    with pytest.raises(ValueError):
        load_reference_weights(m2, PREFIX)


def test_bad_magic_rejected(tmp_path):
    from multihop_offload_amd.utils.tfckpt import read_bundle
    p = tmp_path / "fake.ckpt"
    (tmp_path / "fake.ckpt.index").write_bytes(b"\x00" * 64)
    with pytest.raises(ValueError):
        read_bundle(str(p))


def test_write_bundle_roundtrip(tmp_path):
    """Writer → reader round-trip: keys, shapes, dtypes, values exact."""
    import numpy as np
    from multihop_offload_amd.utils import tfckpt
    tensors = {
        "layer_with_weights-0/kernel/.ATTRIBUTES/VARIABLE_VALUE":
            np.random.RandomState(0).randn(2, 4, 32),
        "layer_with_weights-0/bias/.ATTRIBUTES/VARIABLE_VALUE":
            np.random.RandomState(1).randn(32),
        "small_f32": np.arange(6, dtype=np.float32).reshape(2, 3),
    }
    prefix = str(tmp_path / "cp-0000.ckpt")
    tfckpt.write_bundle(prefix, tensors)
    got = tfckpt.read_bundle(prefix)
    assert set(got) == set(tensors)
    for k, v in tensors.items():
        assert got[k].dtype == np.asarray(v).dtype
        assert np.array_equal(got[k], v)


def test_save_reference_weights_roundtrip(tmp_path):
    """ChebConvStack → TF bundle → load_reference_weights equality, via
    the reference's key layout and the `checkpoint` manifest."""
    import numpy as np
    import torch
    from multihop_offload_amd.models.chebconv import ChebConvStack
    from multihop_offload_amd.utils import tfckpt
    m = ChebConvStack(K=2, dtype=torch.float64, seed=11)
    prefix = str(tmp_path / "cp-0042.ckpt")
    tfckpt.save_reference_weights(m, prefix)
    assert (tmp_path / "checkpoint").read_text().startswith(
        'model_checkpoint_path: "cp-0042.ckpt"')
    m2 = ChebConvStack(K=2, dtype=torch.float64, seed=99)
    tfckpt.load_reference_weights(m2, prefix)
    for p1, p2 in zip(m.parameters(), m2.parameters()):
        assert torch.equal(p1, p2)


def test_written_bundle_crc_matches_tf_convention(tmp_path):
    """Tensor-entry crc32c uses TF's masked CRC-32C (BundleEntryProto
    field 6) — checked against a hand-computed known vector."""
    from multihop_offload_amd.utils.tfckpt import _crc32c, _masked_crc32c
    # RFC 3720 CRC-32C test vector: 32 zero bytes -> 0x8A9136AA
    assert _crc32c(b"\x00" * 32) == 0x8A9136AA
    assert _crc32c(b"123456789") == 0xE3069283
    m = _masked_crc32c(b"123456789")
    c = 0xE3069283
    assert m == (((c >> 15) | (c << 17)) + 0xA282EAD8) & 0xFFFFFFFF


def test_rewrite_reference_checkpoint_equal(tmp_path):
    """Read the reference's shipped TF bundle, re-write it with our
    writer, re-read: tensors identical (format-level TF round-trip)."""
    import numpy as np
    from multihop_offload_amd.utils import tfckpt
    import os
    prefix = ("/root/reference/model/model_ChebConv_BAT950_a5_c5_ACO_agent/"
              "cp-0000.ckpt")
    if not os.path.exists(prefix + ".index"):
        import pytest
        pytest.skip("reference checkpoint not mounted")
    ref = {k: v for k, v in tfckpt.read_bundle(prefix).items()}
    out = str(tmp_path / "cp-0000.ckpt")
    tfckpt.write_bundle(out, ref)
    back = tfckpt.read_bundle(out)
    assert set(back) == set(ref)
    for k in ref:
        assert np.array_equal(back[k], ref[k]), k
