"""LearnedDict family semantics + pickle checkpoint compatibility."""

import io
import os
import pickle

import pytest
import torch

from sparse_coding_amd.models.learned_dict import (
    Identity,
    IdentityPositive,
    IdentityReLU,
    RandomDict,
    ReverseSAE,
    Rotation,
    TiedSAE,
    UntiedSAE,
    normalize_rows,
)


def test_normalize_rows():
    w = torch.randn(8, 4) * 5
    n = normalize_rows(w)
    assert torch.allclose(torch.norm(n, dim=-1), torch.ones(8), atol=1e-6)
    # tiny rows: clamped denominator, not NaN
    w2 = torch.zeros(2, 4)
    assert not normalize_rows(w2).isnan().any()


def test_untied_sae_encode_decode():
    torch.manual_seed(0)
    enc = torch.randn(16, 8)
    dec = torch.randn(16, 8)
    bias = torch.randn(16)
    sae = UntiedSAE(enc, dec, bias)
    x = torch.randn(5, 8)
    c = sae.encode(x)
    # einsum("nd,bd->bn") + bias, clamped
    ref = torch.clamp(torch.einsum("nd,bd->bn", enc, x) + bias, min=0)
    assert torch.allclose(c, ref, atol=1e-5)
    x_hat = sae.decode(c)
    ref_dec = torch.einsum("nd,bn->bd", normalize_rows(dec), c)
    assert torch.allclose(x_hat, ref_dec, atol=1e-5)
    assert torch.allclose(sae.predict(x), ref_dec, atol=1e-5)


def test_tied_sae_centering_roundtrip():
    torch.manual_seed(1)
    d = 6
    enc = torch.randn(12, d)
    bias = torch.zeros(12)
    # random orthogonal rotation
    q, _ = torch.linalg.qr(torch.randn(d, d))
    trans = torch.randn(d)
    scale = torch.rand(d) + 0.5
    sae = TiedSAE(enc, bias, centering=(trans, q, scale))
    x = torch.randn(7, d)
    assert torch.allclose(sae.uncenter(sae.center(x)), x, atol=1e-5)


def test_identity_variants():
    x = torch.randn(4, 6)
    assert torch.equal(Identity(6).encode(x), x)
    pos = IdentityPositive(6)
    c = pos.encode(x)
    assert c.shape == (4, 12)
    assert (c >= 0).all()
    assert torch.allclose(pos.decode(c), x, atol=1e-6)
    relu = IdentityReLU(6)
    assert torch.equal(relu.encode(x), torch.clamp(x, min=0))


def test_reverse_sae_bias_removal():
    torch.manual_seed(2)
    enc = torch.randn(10, 5)
    bias = torch.rand(10) + 0.1
    sae = ReverseSAE(enc, bias, norm_encoder=True)
    x = torch.randn(3, 5)
    c = sae.encode(x)
    x_hat = sae.decode(c.clone())
    on = c > 0
    c_adj = torch.where(on, c - bias, c)
    assert torch.allclose(x_hat, c_adj @ normalize_rows(enc), atol=1e-5)


def test_pickle_module_path_is_reference_compatible():
    """Checkpoints must unpickle as autoencoders.learned_dict.* (SURVEY §2.3)."""
    sae = TiedSAE(torch.randn(4, 3), torch.zeros(4))
    assert type(sae).__module__ == "autoencoders.learned_dict"
    buf = io.BytesIO()
    torch.save([(sae, {"l1_alpha": 1e-3, "dict_size": 4})], buf)
    buf.seek(0)
    loaded = torch.load(buf, weights_only=False)
    ld, hp = loaded[0]
    assert isinstance(ld, TiedSAE)
    assert hp["dict_size"] == 4
    # raw pickle path string check
    buf2 = io.BytesIO()
    pickle.dump(sae.__class__, buf2)
    assert b"autoencoders" in buf2.getvalue()


def test_rotation_and_random_dict():
    q, _ = torch.linalg.qr(torch.randn(5, 5))
    rot = Rotation(q)
    x = torch.randn(3, 5)
    assert torch.allclose(rot.encode(x), x @ q.T, atol=1e-6)
    rd = RandomDict(5, 9)
    assert rd.encode(x).shape == (3, 9)


def test_pickle_attribute_parity_with_reference():
    """A checkpoint saved by the REFERENCE restores its attribute names onto
    our classes (pickle bypasses __init__), so every attribute a reference
    class assigns must also be what our methods read.  Guard against drift
    by diffing the `self.X =` sets per shared class name."""
    import re

    REF = "/root/reference"
    if not os.path.isdir(REF):
        pytest.skip("reference tree not mounted")

    def class_attrs(path):
        classes, cur = {}, None
        for line in open(path).read().splitlines():
            m = re.match(r"^class (\w+)", line)
            if m:
                cur = m.group(1)
                classes[cur] = set()
            elif cur:
                classes[cur].update(re.findall(r"self\.(\w+)\s*=", line))
        return classes

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    pairs = [
        ("autoencoders/learned_dict.py", "sparse_coding_amd/models/learned_dict.py"),
        ("autoencoders/topk_encoder.py", "sparse_coding_amd/models/topk.py"),
        ("autoencoders/sae_ensemble.py", "sparse_coding_amd/models/sae_signatures.py"),
        ("autoencoders/pca.py", "sparse_coding_amd/models/pca.py"),
        ("autoencoders/ica.py", "sparse_coding_amd/models/ica.py"),
        ("autoencoders/nmf.py", "sparse_coding_amd/models/nmf.py"),
        ("autoencoders/residual_denoising_autoencoder.py", "sparse_coding_amd/models/lista.py"),
    ]
    problems = []
    for ref_rel, our_rel in pairs:
        ref = class_attrs(os.path.join(REF, ref_rel))
        ours = class_attrs(os.path.join(root, our_rel))
        for cls, attrs in ref.items():
            if cls in ours and attrs:
                missing = attrs - ours[cls]
                if missing:
                    problems.append((cls, sorted(missing)))
    assert not problems, problems


def test_reference_saved_checkpoint_simulation():
    """Load a checkpoint AS THE REFERENCE WOULD HAVE SAVED IT: objects built
    via __new__ with exactly the reference's attribute sets (pickle never
    calls __init__), then encode/decode/predict must work."""
    import autoencoders.learned_dict as ald

    d, n = 8, 16
    enc = torch.randn(n, d)
    bias = torch.zeros(n)

    # reference TiedSAE attrs: encoder, encoder_bias, norm_encoder,
    # n_feats, activation_size, center_rot/scale/trans
    obj = ald.TiedSAE.__new__(ald.TiedSAE)
    obj.__dict__.update(dict(
        encoder=enc, encoder_bias=bias, norm_encoder=True,
        n_feats=n, activation_size=d,
        center_rot=torch.eye(d), center_scale=torch.ones(d),
        center_trans=torch.zeros(d),
    ))
    buf = io.BytesIO()
    torch.save([(obj, {"l1_alpha": 1e-3, "dict_size": n})], buf)
    buf.seek(0)
    loaded = torch.load(buf, weights_only=False)
    ld, hp = loaded[0]
    x = torch.randn(4, d)
    c = ld.encode(ld.center(x))
    assert c.shape == (4, n) and (c >= 0).all()
    x_hat = ld.predict(x)
    assert x_hat.shape == x.shape and torch.isfinite(x_hat).all()
    D = ld.get_learned_dict()
    assert torch.allclose(torch.norm(D, dim=-1), torch.ones(n), atol=1e-5)

    # UntiedSAE: encoder, decoder, encoder_bias, n_feats, activation_size
    dec = torch.randn(n, d)
    obj2 = ald.UntiedSAE.__new__(ald.UntiedSAE)
    obj2.__dict__.update(dict(encoder=enc, decoder=dec, encoder_bias=bias,
                              n_feats=n, activation_size=d))
    buf = io.BytesIO()
    torch.save(obj2, buf)
    buf.seek(0)
    ld2 = torch.load(buf, weights_only=False)
    assert torch.isfinite(ld2.predict(x)).all()
