import pytest
import torch
import torch.nn.functional as F

from kfac_pytorch_amd.models import (LSTMLanguageModel, get_cifar_model,
                                     get_imagenet_model, make_bert_base_squad,
                                     make_transformer, vgg16, wrn28_10)


def test_cifar_resnets_forward(seeded):
    for name in ("resnet20", "resnet32", "resnet110"):
        m = get_cifar_model(name)
        y = m(torch.randn(2, 3, 32, 32))
        assert y.shape == (2, 10)


def test_imagenet_resnet50_forward(seeded):
    m = get_imagenet_model("resnet50")
    y = m(torch.randn(1, 3, 64, 64))
    assert y.shape == (1, 1000)
    n_conv = sum(1 for mod in m.modules()
                 if mod.__class__.__name__ == "Conv2d")
    assert n_conv == 53  # 53 convs + 1 fc in resnet50


def test_vgg_wrn_forward(seeded):
    assert vgg16()(torch.randn(1, 3, 32, 32)).shape == (1, 10)
    assert wrn28_10()(torch.randn(1, 3, 32, 32)).shape == (1, 10)


def test_transformer_forward_and_kfac(single_process_comm, seeded):
    import kfac_pytorch_amd as kfac
    vocab = 97
    m = make_transformer(vocab=vocab, d_model=32, nhead=4, num_layers=2,
                         dim_ff=64, max_len=32)
    pre = kfac.KFAC_INV_DP(m, damping=0.01, exclude_vocabulary_size=vocab)
    # vocab-sized generator excluded from K-FAC modules
    assert all(getattr(mod, "out_features", None) != vocab
               for mod in pre.modules)
    src = torch.randint(0, vocab, (2, 7))
    trg = torch.randint(0, vocab, (2, 5))
    logits = m(src, trg)
    assert logits.shape == (2, 5, vocab)
    loss = F.cross_entropy(logits.reshape(-1, vocab), trg.reshape(-1))
    loss.backward()
    pre.step()
    assert all(torch.isfinite(p.grad).all() for p in m.parameters()
               if p.grad is not None)


def test_transformer_greedy_decode(seeded):
    m = make_transformer(vocab=50, d_model=16, nhead=2, num_layers=1,
                         dim_ff=32, max_len=16)
    out = m.greedy_decode(torch.randint(0, 50, (2, 5)), max_len=6)
    assert out.shape[0] == 2 and out.shape[1] <= 6


def test_bert_shape_forward_and_kfac(single_process_comm, seeded):
    import kfac_pytorch_amd as kfac
    m = make_bert_base_squad(vocab_size=211)
    m2 = type(m)(vocab_size=211, d_model=32, nhead=2, num_layers=2,
                 dim_ff=64, max_len=32)
    pre = kfac.KFAC_EIGEN_DP(m2, damping=0.01,
                             exclude_vocabulary_size=211)
    ids = torch.randint(0, 211, (2, 12))
    start, end = m2(ids)
    assert start.shape == (2, 12) and end.shape == (2, 12)
    target = torch.randint(0, 12, (2,))
    loss = F.cross_entropy(start, target) + F.cross_entropy(end, target)
    loss.backward()
    pre.step()
    assert all(torch.isfinite(p.grad).all() for p in m2.parameters()
               if p.grad is not None)


def test_rnn_lm_forward_and_kfac(single_process_comm, seeded):
    import kfac_pytorch_amd as kfac
    m = LSTMLanguageModel(vocab_size=120, emb=16, hidden=16, layers=1)
    pre = kfac.KFAC_EIGEN_DP(m, damping=0.01)
    x = torch.randint(0, 120, (2, 9))
    logits, _ = m(x)
    assert logits.shape == (2, 9, 120)
    loss = F.cross_entropy(logits.reshape(-1, 120),
                           torch.randint(0, 120, (18,)))
    loss.backward()
    pre.step()


def test_seq_mean_keeps_factor_dims(single_process_comm, seeded):
    """3-D activations/grads fold to d x d factors regardless of seq len
    (the reference's entire long-context strategy)."""
    import kfac_pytorch_amd as kfac
    import torch.nn as nn

    class M(nn.Module):
        def __init__(self):
            super().__init__()
            self.fc = nn.Linear(8, 6)

        def forward(self, x):
            return self.fc(x)

    m = M()
    pre = kfac.KFAC_EIGEN_DP(m, damping=0.01)
    for seq in (5, 37):
        x = torch.randn(3, seq, 8)
        m.zero_grad(set_to_none=False)
        m(x).sum().backward()
        pre.step()
        assert pre.m_A[m.fc].shape == (9, 9)   # 8 + bias
        assert pre.m_G[m.fc].shape == (6, 6)


def test_imagenet_extra_models_forward(seeded):
    """The reference trainer's remaining model families
    (examples/pytorch_imagenet_resnet.py:235-258) build and forward."""
    for name, size in [("densenet121", 64), ("mobilenetv2", 64),
                       ("inceptionv3", 128), ("inceptionv4", 128),
                       ("vgg16", 64)]:
        m = get_imagenet_model(name, num_classes=13)
        y = m(torch.randn(1, 3, size, size))
        assert y.shape == (1, 13), name


def test_inceptionv4_kfac_hooks(single_process_comm, seeded):
    """Inception-v4 is a BASELINE efficiency config (batch.sh:30);
    K-FAC must hook every conv/linear and step through it."""
    import kfac_pytorch_amd as kfac
    m = get_imagenet_model("inceptionv4", num_classes=5)
    pre = kfac.get_kfac_module("eigen_dp")(m, damping=0.003)
    assert len(pre.modules) > 100
    loss = F.cross_entropy(m(torch.randn(2, 3, 128, 128)),
                           torch.tensor([0, 1]))
    loss.backward()
    pre.step()
    assert all(torch.isfinite(p.grad).all() for p in m.parameters()
               if p.grad is not None)


def test_transformer_beam_decode():
    """Beam search (the reference's Translator analog): returns a
    bos-prefixed sequence, is deterministic, and its length-normalized
    score is at least the greedy path's."""
    import torch
    from kfac_pytorch_amd.models.transformer import Seq2SeqTransformer
    torch.manual_seed(0)
    m = Seq2SeqTransformer(src_vocab=50, trg_vocab=50, d_model=32,
                           nhead=4, num_layers=1, dim_ff=64)
    m.eval()
    src = torch.randint(3, 50, (1, 7))
    out1 = m.beam_decode(src, beam_size=4, max_len=12)
    out2 = m.beam_decode(src, beam_size=4, max_len=12)
    assert torch.equal(out1, out2)
    assert out1[0].item() == 1  # bos
    assert out1.dim() == 1 and 1 <= out1.numel() <= 12
    g = m.greedy_decode(src, max_len=12)[0]
    def score(seq):
        if seq.numel() < 2:
            return -1e9
        logits = m(src, seq[:-1].unsqueeze(0))
        lp = torch.log_softmax(logits[0].float(), -1)
        s = sum(float(lp[t, seq[t + 1]]) for t in range(seq.numel() - 1))
        return s / (seq.numel() ** 0.6)
    assert score(out1) >= score(g) - 1e-4


def test_vit_kfac_step():
    """ViT (beyond-reference zoo entry): every parameterized layer is
    Linear/Conv2d, the full model preconditions end to end."""
    import torch
    import torch.nn.functional as F
    import torch.distributed as dist
    import kfac_pytorch_amd as kfac
    import kfac_pytorch_amd.parallel.comm as comm_mod
    from tests.conftest import free_port
    if not dist.is_initialized():
        dist.init_process_group(
            "gloo", init_method=f"tcp://127.0.0.1:{free_port()}",
            world_size=1, rank=0)
    comm_mod.reset()
    comm_mod.init("Torch")
    from kfac_pytorch_amd.models import get_imagenet_model
    torch.manual_seed(0)
    m = get_imagenet_model("vit_tiny", num_classes=10)
    # shrink for CPU: 32px, patch 16 -> 4 tokens
    from kfac_pytorch_amd.models.vit import VisionTransformer
    m = VisionTransformer(image_size=32, patch=16, dim=48, depth=2,
                          heads=3, num_classes=10)
    pre = kfac.KFAC_EIGEN_DP(m, damping=0.01)
    x = torch.randn(4, 3, 32, 32)
    y = torch.randint(0, 10, (4,))
    m.zero_grad(set_to_none=False)
    F.cross_entropy(m(x), y).backward()
    pre.step()
    # patchify conv + 6 linears/block x2 + head = 14 hooked layers
    assert len(pre.modules) == 1 + 6 * 2 + 1
    for p in m.parameters():
        assert torch.isfinite(p.grad).all()
