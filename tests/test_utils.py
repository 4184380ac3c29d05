"""Unit tests: Philox known-answer vectors, comm pack/unpack, tracing."""

import json

import numpy as np

from flake16_framework_amd.utils.philox import (
    bounded_int, philox4x32, u32_to_unit,
)


class TestPhilox:
    def test_random123_known_answer_vectors(self):
        """The framework RNG is standard Philox4x32-10: the published
        Random123 KAT vectors must reproduce exactly."""
        cases = [
            ((0, 0, 0, 0, 0, 0),
             (0x6627E8D5, 0xE169C58D, 0xBC57AC4C, 0x9B00DBD8)),
            ((0xFFFFFFFF,) * 6,
             (0x408F276D, 0x41C83B0E, 0xA20BC7C6, 0x6D5451FD)),
            ((0x243F6A88, 0x85A308D3, 0x13198A2E, 0x03707344,
              0xA4093822, 0x299F31D0),
             (0xD16CFE09, 0x94FDCCEB, 0x5001E420, 0x24126EA1)),
        ]
        for args, expect in cases:
            got = philox4x32(*args)
            assert tuple(int(x) for x in got) == expect

    def test_vectorized_matches_scalar(self):
        c3 = np.arange(64, dtype=np.uint32)
        x0, _, _, _ = philox4x32(1, 2, 3, c3, 7, 9)
        for i in (0, 17, 63):
            s0, _, _, _ = philox4x32(1, 2, 3, np.uint32(i), 7, 9)
            assert int(x0[i]) == int(s0)

    def test_helpers(self):
        u = np.array([0, 2**31, 2**32 - 1], dtype=np.uint32)
        f = u32_to_unit(u)
        # the final float32 cast rounds (2^32-1)/2^32 up to exactly 1.0f —
        # deliberate: both the numpy and HIP sides share this formula
        assert f[0] == 0.0 and 0.49 < f[1] < 0.51 and f[2] <= 1.0
        b = bounded_int(u, 10)
        assert b[0] == 0 and b[2] == 9
        assert (bounded_int(u, 1) == 0).all()


class TestCommPack:
    def test_pack_unpack_roundtrip(self):
        from flake16_framework_amd.configgrid import iter_config_keys
        from flake16_framework_amd.parallel.comm import _pack, _unpack

        cell_order = list(iter_config_keys())[:4]
        projects = ["a", "b", "c"]
        result = {
            cell_order[1]: [0.5, 0.25,
                            {"a": [1, 2, 3, None, None, None],
                             "b": [4, 5, 6, None, None, None],
                             "c": [0, 0, 0, None, None, None]},
                            [5, 7, 9, None, None, None]],
        }
        buf = _pack(result, cell_order, projects)
        out = _unpack(buf, cell_order, projects)

        got = out[cell_order[1]]
        assert got[0] == 0.5 and got[1] == 0.25
        assert got[2]["a"][:3] == [1, 2, 3]
        assert got[3][:3] == [5, 7, 9]
        # PRF recomputed: P = 9/(9+5), R = 9/(9+7)
        assert abs(got[3][3] - 9 / 14) < 1e-12
        assert abs(got[3][4] - 9 / 16) < 1e-12
        # unevaluated cells are absent (presence column), not phantom
        # zero-count entries
        assert cell_order[0] not in out
        assert set(out) == {cell_order[1]}


class TestTrace:
    def test_span_records(self, tmp_path, monkeypatch):
        from flake16_framework_amd.utils import trace
        path = str(tmp_path / "t.jsonl")
        monkeypatch.setattr(trace, "_explicit", False)
        monkeypatch.setenv("FLAKE16_TRACE", path)
        with trace.trace_span("unit", foo=3):
            pass
        rec = json.loads(open(path).read().strip())
        assert rec["name"] == "unit" and rec["foo"] == 3
        assert rec["dur_s"] >= 0

    def test_disabled_is_noop(self, tmp_path, monkeypatch):
        from flake16_framework_amd.utils import trace
        monkeypatch.setattr(trace, "_explicit", False)
        monkeypatch.delenv("FLAKE16_TRACE", raising=False)
        with trace.trace_span("unit"):
            pass  # must not raise or write anywhere
