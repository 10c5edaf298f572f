"""Offline converter: fastai ``learn.export()`` pickle -> MI355X-native
model artifacts (config.json + vocab.json + encoder.pth).

The deployed reference artifact is the exported inference pickle
(``trained_model_22zkdqlr.pkl``, 965 MB) downloaded at pod start
(/root/reference/Issue_Embeddings/flask_app/app.py:20-34). Unpickling it
normally requires the fastai class tree; this converter needs NO fastai:
a stub-class unpickler materializes every non-torch class as a generic
attribute bag (tensors/Parameters deserialize through torch's own
machinery), then the Learner graph is walked structurally:

  * the encoder = the module subtree whose parameter names match the
    fastai AWD_LSTM layout (``encoder.weight``, ``rnns.{l}...``)
  * the vocab  = the first object carrying an ``itos`` list of strings

Usage:
  python scripts/convert_fastai_pkl.py model.pkl out_dir/
  python -m pytest tests/test_pkl_convert.py     # fixture round-trip

The produced directory loads with ``InferenceWrapper(model_path=out_dir)``
(engine/inference.py). Tokenizer caveat (NOTES.md item 5) still applies:
our tokenizer is not token-for-token identical to spacy+mdparse.
"""
from __future__ import annotations

import io
import json
import pickle
import sys
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))


# ---------------------------------------------------------------------------
# stub unpickler

_ALLOWED_PREFIXES = ("torch", "collections", "builtins", "numpy", "copyreg",
                     "functools", "_codecs", "pathlib", "operator")


class _StubBase:
    """Generic attribute bag standing in for any unavailable class."""

    def __init__(self, *args, **kwargs):
        self._ci_args = args
        self._ci_kwargs = kwargs

    def __setstate__(self, state):
        if isinstance(state, dict):
            self.__dict__.update(state)
        elif isinstance(state, tuple) and len(state) == 2 \
                and isinstance(state[1], dict):
            if isinstance(state[0], dict):
                self.__dict__.update(state[0])
            self.__dict__.update(state[1])
        else:
            self.__dict__["_ci_state"] = state

    def __call__(self, *args, **kwargs):  # reduce-by-call patterns
        return _StubBase()


_stub_cache: dict = {}


def _make_stub(module: str, name: str):
    key = (module, name)
    if key not in _stub_cache:
        _stub_cache[key] = type(name, (_StubBase,),
                                {"_ci_module": module, "_ci_name": name})
    return _stub_cache[key]


class _StubUnpickler(pickle.Unpickler):
    def find_class(self, module, name):
        root = module.split(".")[0]
        if root in _ALLOWED_PREFIXES:
            try:
                return super().find_class(module, name)
            except (ImportError, AttributeError):
                pass
        return _make_stub(module, name)


class _StubPickleModule:
    """pickle_module shim for torch.load."""
    Unpickler = _StubUnpickler
    load = staticmethod(lambda f, **kw: _StubUnpickler(f).load())

    @staticmethod
    def loads(data, **kw):
        return _StubUnpickler(io.BytesIO(data)).load()


def load_with_stubs(path):
    return torch.load(path, map_location="cpu",
                      pickle_module=_StubPickleModule, weights_only=False)


# ---------------------------------------------------------------------------
# graph walking

def _module_like(obj) -> bool:
    d = getattr(obj, "__dict__", None)
    return isinstance(d, dict) and ("_parameters" in d or "_modules" in d)


def collect_state_dict(obj, prefix="") -> dict:
    sd = {}
    d = obj.__dict__
    for k, v in (d.get("_parameters") or {}).items():
        if v is not None:
            sd[prefix + k] = v.detach().clone() if torch.is_tensor(v) else v
    for k, v in (d.get("_buffers") or {}).items():
        if v is not None:
            sd[prefix + k] = v.detach().clone() if torch.is_tensor(v) else v
    for k, m in (d.get("_modules") or {}).items():
        if m is not None and _module_like(m):
            sd.update(collect_state_dict(m, prefix + k + "."))
    return sd


def _walk(obj, seen, depth=0):
    """Yield every reachable python object (bounded)."""
    if id(obj) in seen or depth > 14:
        return
    seen.add(id(obj))
    yield obj
    if isinstance(obj, dict):
        it = list(obj.values())
    elif isinstance(obj, (list, tuple, set)):
        it = list(obj)
    elif hasattr(obj, "__dict__") and not torch.is_tensor(obj):
        it = list(obj.__dict__.values())
    else:
        return
    for v in it:
        yield from _walk(v, seen, depth + 1)


def find_encoder_state(root) -> dict:
    """Find the module subtree with the fastai AWD_LSTM encoder layout."""
    best = None
    for obj in _walk(root, set()):
        if not _module_like(obj):
            continue
        sd = collect_state_dict(obj)
        if "encoder.weight" in sd and any(k.startswith("rnns.0.") for k in sd):
            if best is None or len(sd) > len(best):
                best = sd
    if best is None:
        raise ValueError("no AWD_LSTM encoder subtree found in pickle")
    return best


def find_vocab(root):
    for obj in _walk(root, set()):
        itos = getattr(obj, "itos", None) if not isinstance(obj, dict) \
            else obj.get("itos")
        if isinstance(itos, (list, tuple)) and len(itos) > 2 \
                and all(isinstance(s, str) for s in itos[:50]):
            return list(itos)
    return None


def infer_config(sd: dict) -> dict:
    vocab_sz, emb_sz = sd["encoder.weight"].shape
    n_layers = 1 + max(int(k.split(".")[1]) for k in sd
                       if k.startswith("rnns."))
    qrnn = not any("weight_hh_l0_raw" in k for k in sd)
    if qrnn:
        n_hid = sd["rnns.0.weight_raw"].shape[0] // 3 \
            if "rnns.0.weight_raw" in sd else 0
    else:
        n_hid = sd["rnns.0.weight_hh_l0_raw"].shape[1]
    return {"emb_sz": int(emb_sz), "n_hid": int(n_hid),
            "n_layers": int(n_layers), "vocab_sz": int(vocab_sz),
            "qrnn": bool(qrnn), "encoder_file": "encoder.pth"}


def convert(pkl_path, out_dir) -> dict:
    out = Path(out_dir)
    out.mkdir(parents=True, exist_ok=True)
    root = load_with_stubs(pkl_path)
    sd = find_encoder_state(root)
    cfg = infer_config(sd)
    itos = find_vocab(root)
    if itos is None:
        raise ValueError("no vocab (itos) found in pickle")
    if len(itos) != cfg["vocab_sz"]:
        print(f"warning: vocab size {len(itos)} != embedding rows "
              f"{cfg['vocab_sz']}", file=sys.stderr)
    torch.save(sd, out / "encoder.pth")
    (out / "config.json").write_text(json.dumps(cfg, indent=1))
    (out / "vocab.json").write_text(json.dumps(itos))
    return cfg


if __name__ == "__main__":
    if len(sys.argv) != 3:
        raise SystemExit(__doc__)
    cfg = convert(sys.argv[1], sys.argv[2])
    print(json.dumps(cfg))
