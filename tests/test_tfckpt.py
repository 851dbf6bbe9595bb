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
