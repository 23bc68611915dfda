"""Native C++ checkpoint I/O vs torch serialization (both directions).

The reference deployment depends on `model_params.pt` (torch zip format,
predict.py:104 / notebook cell 39); the native reader/writer
(fmda_amd/ops/csrc/checkpoint.cpp) must interoperate byte-level with
torch.save/torch.load including the real BiGRU state_dict layout."""
import collections

import pytest
import torch

from fmda_amd.models import BiGRU
from fmda_amd.ops import load_extension

ext = pytest.importorskip("fmda_amd.ops._fmda_hip")


def _bigru_sd():
    torch.manual_seed(0)
    # the shipped checkpoint architecture: H=8, F=108, C=4, bidirectional
    return BiGRU(8, 108, 4, n_layers=1, spatial_dropout=False).state_dict()


def test_native_save_torch_load(tmp_path):
    sd = _bigru_sd()
    p = str(tmp_path / "m.pt")
    ext.save_state_dict_native(p, list(sd.keys()), list(sd.values()))
    sd2 = torch.load(p)  # weights_only=True default must accept it
    assert list(sd2.keys()) == list(sd.keys())
    for k in sd:
        assert torch.equal(sd[k], sd2[k]), k


def test_torch_save_native_load(tmp_path):
    sd = _bigru_sd()
    p = str(tmp_path / "m.pt")
    torch.save(sd, p)
    got = dict(ext.load_state_dict_native(p))
    assert set(got.keys()) == set(sd.keys())
    for k in sd:
        assert torch.equal(sd[k], got[k]), k


def test_model_loads_native_checkpoint(tmp_path):
    sd = _bigru_sd()
    p = str(tmp_path / "m.pt")
    ext.save_state_dict_native(p, list(sd.keys()), list(sd.values()))
    m = BiGRU(8, 108, 4, n_layers=1, spatial_dropout=False)
    m.load_state_dict(torch.load(p))
    x = torch.randn(2, 5, 108)
    m.eval()
    assert m(x).shape == (2, 4)


def test_mixed_dtypes_roundtrip(tmp_path):
    sd = collections.OrderedDict()
    torch.manual_seed(1)
    sd["a"] = torch.randn(7, 3)
    sd["b"] = torch.randn(11).to(torch.bfloat16)
    sd["c"] = torch.arange(5, dtype=torch.int64)
    sd["d"] = torch.randn(2, 2, 2, dtype=torch.float64)
    p = str(tmp_path / "m.pt")
    ext.save_state_dict_native(p, list(sd.keys()), list(sd.values()))
    got = dict(ext.load_state_dict_native(p))
    sd2 = torch.load(p)
    for k in sd:
        assert torch.equal(sd[k], got[k]), k
        assert torch.equal(sd[k], sd2[k]), k


def test_native_reader_rejects_malformed_files(tmp_path):
    """The C++ reader must raise clean RuntimeErrors (never crash) on
    malformed input: garbage, truncated zip containers, truncated legacy
    pickles."""
    import pytest
    import torch

    from fmda_amd.ops import load_extension
    ext = load_extension()
    if ext is None:
        pytest.skip("extension not built")

    cases = [b"", b"\x00\x01junk" * 16,
             b"PK\x03\x04" + b"\x00" * 20,
             (0x1950a86a20f9469c).to_bytes(8, "little")]
    sd = {"gru.weight_ih_l0": torch.randn(6, 4)}
    zp = tmp_path / "z.pt"
    torch.save(sd, str(zp))
    cases.append(zp.read_bytes()[: zp.stat().st_size // 2])
    lp = tmp_path / "l.pt"
    torch.save(sd, str(lp), _use_new_zipfile_serialization=False)
    cases.append(lp.read_bytes()[: int(lp.stat().st_size * 0.4)])
    cases.append(lp.read_bytes()[: int(lp.stat().st_size * 0.9)])

    for i, data in enumerate(cases):
        p = tmp_path / f"bad_{i}"
        p.write_bytes(data)
        with pytest.raises(RuntimeError):
            ext.load_state_dict_native(str(p))


def test_native_roundtrip_property(tmp_path):
    """Property test: random state dicts (shapes, dtypes, key names) round
    trip through BOTH directions — native writer -> torch.load and
    torch.save -> native reader — bit-exactly."""
    import torch
    from hypothesis import given, settings, strategies as st

    from fmda_amd.ops import load_extension
    ext = load_extension()

    dtypes = [torch.float32, torch.float64, torch.int64, torch.int32,
              torch.float16, torch.bfloat16, torch.bool]

    @settings(max_examples=25, deadline=None)
    @given(st.lists(
        st.tuples(st.text(alphabet="abc_.019", min_size=1, max_size=12),
                  st.lists(st.integers(0, 7), min_size=0, max_size=3),
                  st.integers(0, len(dtypes) - 1)),
        min_size=1, max_size=5, unique_by=lambda t: t[0]))
    def check(spec):
        g = torch.Generator().manual_seed(7)
        sd = {}
        for name, shape, di in spec:
            dt = dtypes[di]
            if dt.is_floating_point:
                t = torch.randn(shape, generator=g).to(dt)
            elif dt == torch.bool:
                t = torch.rand(shape, generator=g) < 0.5
            else:
                t = (torch.rand(shape, generator=g) * 100).to(dt)
            sd[name] = t
        p1 = str(tmp_path / "n.pt")
        ext.save_state_dict_native(p1, list(sd.keys()), list(sd.values()))
        back = dict(torch.load(p1, weights_only=True))
        assert set(back) == set(sd)
        for k in sd:
            assert back[k].dtype == sd[k].dtype
            assert torch.equal(back[k], sd[k]), k
        p2 = str(tmp_path / "t.pt")
        torch.save(sd, p2)
        back2 = dict(ext.load_state_dict_native(p2))
        for k in sd:
            assert torch.equal(back2[k], sd[k]), k

    check()
