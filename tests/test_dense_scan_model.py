"""Host-side model of the dense agg kernel's wave algorithm.

`agg_apply_dense4_body` (rw_amd.hip) splits each 64-lane x RPL-row tile
into lane-local equal-key segments, then closes cross-lane runs with a
segmented inclusive scan over lane suffixes; exactly one commit reaches
the table per (tile, run). The count-star call is carried as a u32 run
length beside the scan (template param CS) instead of an i64 inside it.

This file re-states that algorithm in Python, lane for lane — including
the CS count path and inactive-row boundaries — and fuzzes it against a
direct per-key reduction. It validates the ALGORITHM (commit set + values
+ counts); the binary is validated on hardware by the GPU parity suite.
"""
import numpy as np
import pytest

RPL = 4
LANES = 64

I64_MIN = -(1 << 63)


def comb_max(a, b):
    return max(a, b)


def wave_commits(keys, act):
    """Model one tile: keys/act are [LANES][RPL]; returns list of
    (key, max_value, count) commits. Value column == key column here
    (wlog: the scan combines values the same way regardless of which
    column they load from); max agg + count-star, the q7 shape."""
    commits = []

    # --- per-lane segment pass ---
    lane_state = []
    for lane in range(LANES):
        cur_key = None
        cur_v = None
        cur_n = 0
        pre = None  # (key, v, n)
        has_bnd = False
        any_row = False
        first_key = None
        for r in range(RPL):
            if not act[lane][r]:
                if cur_key is not None:
                    if pre is None:
                        pre = (cur_key, cur_v, cur_n)
                    else:
                        commits.append((cur_key, cur_v, cur_n))
                    has_bnd = True
                    cur_key = None
                continue
            any_row = True
            k = keys[lane][r]
            if first_key is None and r == 0:
                first_key = k
            if cur_key is not None and k == cur_key:
                cur_v = comb_max(cur_v, k)
                cur_n += 1
            else:
                if cur_key is not None:
                    if pre is None:
                        pre = (cur_key, cur_v, cur_n)
                    else:
                        commits.append((cur_key, cur_v, cur_n))
                    has_bnd = True
                cur_key, cur_v, cur_n = k, k, 1
        have_suf = cur_key is not None
        suf = (cur_key, cur_v, cur_n) if have_suf else (None, I64_MIN, 0)
        if pre is not None and have_suf:
            has_bnd = True
        last_key = suf[0] if have_suf else (pre[0] if pre else None)
        my_first = (pre[0] if pre else suf[0]) if any_row else None
        first_at_row0 = any_row and act[lane][0] and keys[lane][0] == my_first
        lane_state.append(dict(any=any_row, bnd=has_bnd, pre=pre, suf=suf,
                               last=last_key, first=my_first,
                               first_at_row0=first_at_row0))

    # --- inter-lane continuity ---
    cont = [False] * LANES
    for lane in range(1, LANES):
        s, p = lane_state[lane], lane_state[lane - 1]
        cont[lane] = (s["any"] and p["any"] and s["first_at_row0"]
                      and s["first"] == p["last"])

    # --- segmented scan over suffixes (value + count) ---
    scan_head = [not lane_state[l]["any"] or lane_state[l]["bnd"]
                 or not cont[l] for l in range(LANES)]
    incl = [None] * LANES
    for lane in range(LANES):
        run_start = lane
        while run_start > 0 and not scan_head[run_start]:
            run_start -= 1
        v, n = I64_MIN, 0
        for l in range(run_start, lane + 1):
            sv = lane_state[l]["suf"]
            v = comb_max(v, sv[1])
            n += sv[2]
        incl[lane] = (v, n)

    # --- commits ---
    for lane in range(LANES):
        s = lane_state[lane]
        if not s["any"]:
            continue
        if cont[lane] and s["bnd"]:
            pv, pn = incl[lane - 1]
            k, v, n = s["pre"]
            commits.append((k, comb_max(pv, v), pn + n))
        elif not cont[lane] and s["bnd"] and s["pre"] is not None:
            commits.append(s["pre"])
        next_cont = lane < LANES - 1 and cont[lane + 1]
        if s["suf"][0] is not None and not next_cont:
            v, n = incl[lane]
            commits.append((s["suf"][0], v, n))
    return commits


def reduce_commits(commits):
    out = {}
    for k, v, n in commits:
        if k in out:
            out[k] = (comb_max(out[k][0], v), out[k][1] + n)
        else:
            out[k] = (v, n)
    return out


def ground_truth(keys, act):
    out = {}
    for lane in range(LANES):
        for r in range(RPL):
            if not act[lane][r]:
                continue
            k = keys[lane][r]
            if k in out:
                out[k] = (comb_max(out[k][0], k), out[k][1] + 1)
            else:
                out[k] = (k, 1)
    return out


@pytest.mark.parametrize("seed", range(8))
def test_dense_scan_model_fuzz(seed):
    rng = np.random.default_rng(seed)
    for trial in range(4000):
        # mix of regimes: long monotone runs (q7), short runs, random keys,
        # and ragged tails / hidden rows
        style = trial % 4
        n = LANES * RPL
        if style == 0:  # few giant runs (monotone windows)
            k = np.sort(rng.integers(0, 3, n))
        elif style == 1:  # medium runs
            k = np.sort(rng.integers(0, 20, n))
        elif style == 2:  # random (worst case, 1-row segments)
            k = rng.integers(0, 1 << 40, n)
        else:  # runs with repeats scattered (non-adjacent same key)
            k = rng.integers(0, 8, n)
        keys = k.reshape(LANES, RPL).tolist()
        if trial % 3 == 0:
            act = np.ones((LANES, RPL), bool)
            tail = int(rng.integers(0, n))  # ragged tail: rows >= tail off
            act.reshape(-1)[tail:] = False
        else:
            act = rng.random((LANES, RPL)) > 0.1
        act = act.tolist()
        got = reduce_commits(wave_commits(keys, act))
        want = ground_truth(keys, act)
        assert got == want, f"seed={seed} trial={trial}"
