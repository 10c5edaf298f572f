"""Reference-artifact bridge: fastai .pkl -> native
artifacts, WITHOUT fastai installed.

The fixture builds a Learner-shaped object graph whose classes live in
fake ``fastai.*`` modules, pickles it with torch.save, then DELETES the
fake modules — so the converter must succeed with the class tree absent,
exactly like the real 965 MB export in an image without fastai. The
golden-vector check embeds 10 synthetic issues through the converted
artifacts and through the original in-memory encoder: vectors must match.
"""
import json
import sys
import types

import numpy as np
import pytest
import torch
from torch import nn

from code_intelligence_amd.models.awd_lstm import AWDLSTM
from code_intelligence_amd.text.tokenizer import Vocab, defaults_specials


def _mk_fake_class(modname, name, base=object):
    parts = modname.split(".")
    for i in range(1, len(parts) + 1):
        mn = ".".join(parts[:i])
        if mn not in sys.modules:
            sys.modules[mn] = types.ModuleType(mn)
    cls = type(name, (base,), {"__module__": modname})
    setattr(sys.modules[modname], name, cls)
    return cls


def _drop_fake_modules():
    for mn in list(sys.modules):
        if mn == "fastai" or mn.startswith("fastai."):
            del sys.modules[mn]


@pytest.fixture
def fake_pkl(tmp_path):
    torch.manual_seed(31)
    vocab_sz, emb, hid, layers = 220, 32, 48, 2
    words = [f"tok{i}" for i in range(vocab_sz - len(defaults_specials))]
    itos = defaults_specials + words
    model = AWDLSTM(vocab_sz=vocab_sz, emb_sz=emb, n_hid=hid,
                    n_layers=layers)

    FVocab = _mk_fake_class("fastai.text.transform", "Vocab")
    FSeq = _mk_fake_class("fastai.text.models", "SequentialRNN",
                          nn.Sequential)
    FLearner = _mk_fake_class("fastai.basic_train", "Learner")
    FDataState = _mk_fake_class("fastai.data_block", "LabelLists")

    fv = FVocab()
    fv.itos = itos
    seq = FSeq(model.encoder, model.decoder)
    learner = FLearner()
    learner.model = seq
    data = FDataState()
    data.vocab = fv
    learner.data = data
    path = tmp_path / "trained_model_fixture.pkl"
    torch.save(learner, path)
    ref_encoder_sd = {k: v.detach().clone()
                      for k, v in model.encoder.state_dict().items()}
    _drop_fake_modules()
    yield path, itos, ref_encoder_sd, (vocab_sz, emb, hid, layers)
    _drop_fake_modules()


def test_convert_without_fastai(fake_pkl, tmp_path):
    path, itos, ref_sd, (V, E, H, L) = fake_pkl
    assert "fastai" not in sys.modules
    from scripts.convert_fastai_pkl import convert
    out = tmp_path / "artifacts"
    cfg = convert(path, out)
    assert cfg["emb_sz"] == E and cfg["n_hid"] == H and cfg["n_layers"] == L
    assert not cfg["qrnn"]
    assert json.loads((out / "vocab.json").read_text()) == itos
    sd = torch.load(out / "encoder.pth", weights_only=True)
    for k, v in ref_sd.items():
        assert k in sd, k
        assert torch.equal(sd[k], v), k


def test_converted_artifacts_golden_vectors(fake_pkl, tmp_path):
    """InferenceWrapper over the converted directory produces the same
    embedding vectors as the original in-memory encoder."""
    path, itos, ref_sd, (V, E, H, L) = fake_pkl
    from scripts.convert_fastai_pkl import convert
    from code_intelligence_amd.engine.inference import InferenceWrapper
    from code_intelligence_amd.data.synthetic import synthetic_issue_texts
    out = tmp_path / "artifacts"
    convert(path, out)

    w_conv = InferenceWrapper(model_path=str(out), device="cpu",
                              dtype=torch.float32)
    model = AWDLSTM(vocab_sz=V, emb_sz=E, n_hid=H, n_layers=L)
    model.encoder.load_state_dict(ref_sd)
    w_ref = InferenceWrapper(encoder=model.encoder, vocab=Vocab(itos),
                             device="cpu", dtype=torch.float32)
    issues = synthetic_issue_texts(10, seed=5)
    for d in issues:
        t_conv = w_conv.process_dict(d)["text"]
        t_ref = w_ref.process_dict(d)["text"]
        assert t_conv == t_ref
        a = w_conv.get_pooled_features(t_conv).numpy()
        b = w_ref.get_pooled_features(t_ref).numpy()
        assert a.shape == (1, 3 * E)
        np.testing.assert_allclose(a, b, atol=1e-6)


def test_convert_errors_without_encoder_or_vocab(tmp_path):
    """Malformed pickles raise clear ValueErrors instead of writing
    partial artifacts."""
    import pytest as _pytest
    from scripts.convert_fastai_pkl import convert
    torch.save({"just": "a dict", "tensor": torch.zeros(3)},
               tmp_path / "noenc.pkl")
    with _pytest.raises(ValueError, match="no AWD_LSTM encoder"):
        convert(tmp_path / "noenc.pkl", tmp_path / "out1")

    enc_only = AWDLSTM(vocab_sz=64, emb_sz=16, n_hid=24, n_layers=2).encoder
    torch.save({"model": enc_only}, tmp_path / "novocab.pkl")
    with _pytest.raises(ValueError, match="no vocab"):
        convert(tmp_path / "novocab.pkl", tmp_path / "out2")
