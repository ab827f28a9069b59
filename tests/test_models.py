"""Model zoo: architecture parity (param counts from BASELINE.md) and
forward shapes."""
import pytest
import torch

from blades_amd.models import (MLP, cct_2_3x2_32, get_model, num_params,
                               resnet18, wide_resnet28_10)


def test_mlp_param_count_exact():
    # reference MLP is 784-64-128-10 → 59,850 params
    assert num_params(MLP()) == 59850


def test_resnet18_param_count():
    n = num_params(resnet18())
    assert abs(n - 11.2e6) / 11.2e6 < 0.01  # ≈11.2M


def test_wrn28_10_param_count():
    n = num_params(wide_resnet28_10())
    assert abs(n - 36.5e6) / 36.5e6 < 0.01  # ≈36.5M


def test_cct_param_count():
    n = num_params(cct_2_3x2_32())
    assert abs(n - 284e3) / 284e3 < 0.02  # ≈284k


@pytest.mark.parametrize("name,shape,classes", [
    ("mlp", (2, 1, 28, 28), 10),
    ("resnet18", (2, 3, 32, 32), 10),
    ("wrn28_10", (2, 3, 32, 32), 10),
    ("cct", (2, 3, 32, 32), 10),
])
def test_forward_shapes(name, shape, classes):
    m = get_model(name)
    out = m(torch.randn(*shape))
    assert out.shape == (shape[0], classes)


def test_cct_sequence_length_is_64():
    m = cct_2_3x2_32()
    assert m.seq_len == 64  # 32x32 through two stride-2 pools → 8x8


def test_norm_variants():
    for norm in ["batch", "batch-local", "group"]:
        m = resnet18(norm=norm)
        out = m(torch.randn(2, 3, 32, 32))
        assert out.shape == (2, 10)


def test_num_classes_knob():
    m = wide_resnet28_10(num_classes=100)
    assert m(torch.randn(2, 3, 32, 32)).shape == (2, 100)


def test_backward_runs():
    for name in ["mlp", "resnet18", "cct"]:
        m = get_model(name)
        shape = (2, 1, 28, 28) if name == "mlp" else (2, 3, 32, 32)
        loss = torch.nn.functional.cross_entropy(
            m(torch.randn(*shape)), torch.tensor([1, 2]))
        loss.backward()
        assert all(p.grad is not None for p in m.parameters()
                   if p.requires_grad)


def test_text_cct_variants():
    """Masked text models (reference cctnets/text/): forward+backward with
    padding masks; masked positions must not affect the logits."""
    import torch

    from blades_amd.models.text_cct import (text_cct_2, text_transformer_2)

    for fn in (text_cct_2, text_transformer_2):
        m = fn(seq_len=32, vocab_size=100, num_classes=4,
               word_embedding_dim=64)
        m.eval()
        ids = torch.randint(1, 100, (3, 32))
        ids[:, 20:] = 0  # padding
        out = m(ids)
        assert out.shape == (3, 4)
        # changing PADDED ids must not change the output
        ids2 = ids.clone()
        ids2[:, 25:] = 0  # same mask, same content
        assert torch.allclose(m(ids2), out, atol=1e-6)
        m.train()
        loss = torch.nn.functional.cross_entropy(
            m(ids), torch.randint(0, 4, (3,)))
        loss.backward()


def test_cct_load_pretrained_resize(tmp_path):
    """Local-checkpoint loading with pos-embed resize + head reset
    (reference: cctnets pretrained-URL path, no egress here)."""
    import torch

    from blades_amd.models.cct import CCT, load_pretrained

    src = CCT(img_size=32, num_classes=10)
    p = str(tmp_path / "cct.pt")
    torch.save(src.state_dict(), p)

    # different image size (pos-embed resize) AND class count (head reset)
    dst = CCT(img_size=64, num_classes=5)
    load_pretrained(dst, p)
    out = dst(torch.randn(2, 3, 64, 64))
    assert out.shape == (2, 5)
    # same-shape load stays bitwise
    dst2 = CCT(img_size=32, num_classes=10)
    load_pretrained(dst2, p)
    assert torch.equal(dst2.state_dict()["fc.weight"],
                       src.state_dict()["fc.weight"])
