#!/usr/bin/env python3
"""Extract the reference's KIP-9 storage-mass / plurality / cofactor golden
vectors (data, not code) into tests/golden/mass.json.

Runs ONLY in the build container where /root/reference is mounted; the JSON
output is committed and is what travels to the GPU box.

Sources (all inside the reference's own test code, consensus/core/src/mass/mod.rs):
  - verify_utxo_plurality_limits   (:531-566)  plurality(spk_len, has_covenant) cases
  - test_storage_mass_pluralities  (:631-748)  super-entry equality cases
  - test_storage_mass              (:758-820)  concrete expected storage-mass values
  - test_mass_cofactors            (:861-925)  limits -> cofactors + normalized_max invariant
  - test_normalized_max_ranking    (:927-953)  exact normalized values

The expected values hand-transcribed below are cross-checked against an
independent Python restatement of calc_storage_mass (mass/mod.rs:439-514)
before the JSON is written, so a transcription slip fails here, not in CI.
"""
import json
import math
import os

OUT = os.path.dirname(os.path.abspath(__file__))

SOMPI = 100_000_000            # constants.rs:23
STORM = SOMPI * 10_000         # STORAGE_MASS_PARAMETER, constants.rs:26 = 1e12
C12 = 10**12


def plurality(spk_len, has_cov):
    """utxo_plurality, mass/mod.rs:83-105: ceil((63 + len + 32?has_cov)/100)."""
    return -(-(63 + spk_len + (32 if has_cov else 0)) // 100)


def calc_storage_mass(ins, outs, storm):
    """Python restatement of calc_storage_mass (mass/mod.rs:439-514).
    ins/outs: lists of (amount, spk_len, has_cov). Returns mass or None."""
    outs_p, harm_outs = 0, 0
    for amount, spk_len, has_cov in outs:
        p = plurality(spk_len, has_cov)
        outs_p += p
        harm_outs += storm * p * p // amount
        if harm_outs >= 2**64:
            return None
    if outs_p == 1:
        relaxed = True
    elif len(ins) > 2:
        relaxed = False
    else:
        ins_p = sum(plurality(sl, hc) for _, sl, hc in ins)
        relaxed = ins_p == 1 or (outs_p == 2 and ins_p == 2)
    if relaxed:
        harm_ins = 0
        for amount, spk_len, has_cov in ins:
            p = plurality(spk_len, has_cov)
            harm_ins = min(2**64 - 1, harm_ins + storm * p * p // amount)
        return max(0, harm_outs - harm_ins)
    ins_p = sum(plurality(sl, hc) for _, sl, hc in ins)
    sum_ins = sum(a for a, _, _ in ins)
    mean = max(1, sum_ins // ins_p)
    arith = min(2**64 - 1, ins_p * (storm // mean))
    return max(0, harm_outs - arith)


def amounts(vals):
    return [(v, 0, False) for v in vals]


# ---- verify_utxo_plurality_limits (:546-566): direct plurality assertions ----
PLURALITY_CASES = [
    # (spk_len, has_covenant, expected)
    (0, False, 1),
    (100 - 63, False, 1),        # 37-byte spk still one unit
    (100 - 63 + 1, False, 2),    # 38 bytes tips into the second unit
    (100 - 63, True, 2),         # covenant id adds 32 bytes
    (2 * 100 - 63 - 32, False + True and True, 2),  # see fixup below
    (2 * 100 - 63 - 32 + 1, True, 3),
]
# the 5th case is (2*UNIT - CONST - COVENANT, has_covenant=True) == 2
PLURALITY_CASES[4] = (2 * 100 - 63 - 32, True, 2)

# ---- test_storage_mass (:758-820): concrete values, all with spk_len 0 ----
BV = 10_000 * SOMPI
STORAGE_MASS_CASES = [
    # (ins amounts, outs amounts, storm, expected)
    ([100, 200, 300], [300, 300], C12, 0),
    ([100, 200, 300], [50, 550], C12,
     C12 // 50 + C12 // 550 - 3 * (C12 // 200)),
    ([BV, BV, BV * 2], [BV] * 4, STORM, 4),
    ([BV, BV, BV * 2], [10 * SOMPI, BV, BV, BV], STORM, 1003),
    ([BV + 4, BV, BV * 2], [BV + 1] * 4, STORM, 0),
    ([100, 200], [50, 250], C12, 9_000_000_000),
    ([100, 200], [100, 200], C12, 0),
    ([100, 200], [50], C12, 5_000_000_000),
]

# ---- test_storage_mass_pluralities (:631-748) ----
# tx1 all-plurality-1 vs tx2 with one entry's spk overridden so its plurality
# is `plur` (spk_len = (plur-1)*100, generate_script_for_plurality :751-755).
# Masses must be equal and nonzero. storm = 1e12 in every case.
PLURALITY_EQ_CASES = [
    # (name, ins1, outs1, ins2, outs2, index, plur, override_output)
    ("3:4 in1 p2", [300, 200, 200], [200, 200, 200, 100],
     [300, 400], [200, 200, 200, 100], 1, 2, False),
    ("2:3 out1 p2", [350, 400], [300, 200, 200],
     [350, 400], [300, 400], 1, 2, True),
    ("1:2 out0 p2", [500], [200, 200], [500], [400], 0, 2, True),
    ("1:3 out1 p2", [1000], [200, 200, 200], [1000], [200, 400], 1, 2, True),
    ("1:3 out1 p2 kas", [1000 * SOMPI], [200 * SOMPI] * 3,
     [1000 * SOMPI], [200 * SOMPI, 400 * SOMPI], 1, 2, True),
    ("1:2 out0 p2 kas", [1000 * SOMPI], [200 * SOMPI] * 2,
     [1000 * SOMPI], [400 * SOMPI], 0, 2, True),
    ("2:2 out0 p2 kas", [350 * SOMPI, 500 * SOMPI], [200 * SOMPI] * 2,
     [350 * SOMPI, 500 * SOMPI], [400 * SOMPI], 0, 2, True),
    ("4:6 out3 p3 kas",
     [350 * SOMPI, 500 * SOMPI, 350 * SOMPI, 500 * SOMPI],
     [200 * SOMPI, 200 * SOMPI, 400 * SOMPI, 250 * SOMPI, 250 * SOMPI, 250 * SOMPI],
     [350 * SOMPI, 500 * SOMPI, 350 * SOMPI, 500 * SOMPI],
     [200 * SOMPI, 200 * SOMPI, 400 * SOMPI, 750 * SOMPI], 3, 3, True),
]

# ---- test_mass_cofactors (:861-925) ----
COFACTOR_CASES = [
    # (storage_limit, compute_limit, transient_limit,
    #  expected_storage_cofactor, expected_transient_cofactor, expected_reference)
    (500_000, 500_000, 500_000, 1.0, 1.0, 500_000),
    (1_000_000, 500_000, 250_000, 0.5, 2.0, 500_000),
    (123_456, 78_901, 45_678, 78_901 / 123_456, 78_901 / 45_678, 78_901),
    (333_333, 77_777, 12_345, 77_777 / 333_333, 77_777 / 12_345, 77_777),
    (1_048_575, 524_287, 262_143, 524_287 / 1_048_575, 524_287 / 262_143, 524_287),
]

# ---- test_normalized_max_ranking (:927-953): limits (1M, 500K, 250K) ----
NORMALIZED_MAX_CASES = [
    # (storage_mass, compute_mass, transient_mass, Ls, Lc, Lt, expected)
    (500_000, 0, 0, 1_000_000, 500_000, 250_000, 250_000),
    (0, 300_000, 0, 1_000_000, 500_000, 250_000, 300_000),
    (0, 0, 200_000, 1_000_000, 500_000, 250_000, 400_000),
]


def normalized_max(s, c, t, ls, lc, lt):
    """Mass::normalized_max (mass/mod.rs:298-308) + cofactors (:258-265)."""
    cs, ct = lc / ls, lc / lt
    return max(math.ceil(s * cs), c, math.ceil(t * ct))


def main():
    # cross-check every transcription against the restatement
    for spk_len, has_cov, exp in PLURALITY_CASES:
        assert plurality(spk_len, has_cov) == exp, (spk_len, has_cov)
    for ins, outs, storm, exp in STORAGE_MASS_CASES:
        got = calc_storage_mass(amounts(ins), amounts(outs), storm)
        assert got == exp, (ins, outs, storm, got, exp)
    eq_expected = []
    for name, i1, o1, i2, o2, idx, plur, over in PLURALITY_EQ_CASES:
        assert sum(i1) >= sum(o1) and sum(i2) >= sum(o2), name
        ins2, outs2 = amounts(i2), amounts(o2)
        spk = (plur - 1) * 100
        if over:
            outs2[idx] = (outs2[idx][0], spk, False)
        else:
            ins2[idx] = (ins2[idx][0], spk, False)
        m1 = calc_storage_mass(amounts(i1), amounts(o1), C12)
        m2 = calc_storage_mass(ins2, outs2, C12)
        assert m1 == m2 and m1 not in (0, None), (name, m1, m2)
        eq_expected.append(m1)
    # cofactor invariant: a tx filling one dimension to its limit normalizes
    # to the reference (mass/mod.rs:905-921)
    for ls, lc, lt, cs, ct, ref in COFACTOR_CASES:
        assert abs(lc / ls - cs) < 1e-10 and abs(lc / lt - ct) < 1e-10
        assert normalized_max(ls, 0, 0, ls, lc, lt) == ref
        assert normalized_max(0, lc, 0, ls, lc, lt) == ref
        assert normalized_max(0, 0, lt, ls, lc, lt) == ref
    for s, c, t, ls, lc, lt, exp in NORMALIZED_MAX_CASES:
        assert normalized_max(s, c, t, ls, lc, lt) == exp

    out = {
        "source": "consensus/core/src/mass/mod.rs tests (:531-953)",
        "storm_param": STORM,
        "plurality": [
            {"spk_len": sl, "has_covenant": hc, "expected": exp}
            for sl, hc, exp in PLURALITY_CASES
        ],
        "storage_mass": [
            {"ins": ins, "outs": outs, "storm": storm, "expected": exp}
            for ins, outs, storm, exp in STORAGE_MASS_CASES
        ],
        "plurality_equalities": [
            {
                "name": name, "ins1": i1, "outs1": o1, "ins2": i2, "outs2": o2,
                "override_index": idx, "override_spk_len": (plur - 1) * 100,
                "override_output": over, "storm": C12, "expected": exp,
            }
            for (name, i1, o1, i2, o2, idx, plur, over), exp
            in zip(PLURALITY_EQ_CASES, eq_expected)
        ],
        "cofactors": [
            {"storage": ls, "compute": lc, "transient": lt,
             "storage_cofactor": cs, "transient_cofactor": ct, "reference": ref}
            for ls, lc, lt, cs, ct, ref in COFACTOR_CASES
        ],
        "normalized_max": [
            {"storage_mass": s, "compute_mass": c, "transient_mass": t,
             "limits": [ls, lc, lt], "expected": exp}
            for s, c, t, ls, lc, lt, exp in NORMALIZED_MAX_CASES
        ],
    }
    with open(os.path.join(OUT, "mass.json"), "w") as f:
        json.dump(out, f, indent=1)
    print(f"mass.json: {len(PLURALITY_CASES)} plurality, "
          f"{len(STORAGE_MASS_CASES)} storage-mass, "
          f"{len(PLURALITY_EQ_CASES)} equality, {len(COFACTOR_CASES)} cofactor, "
          f"{len(NORMALIZED_MAX_CASES)} normalized-max cases")


if __name__ == "__main__":
    main()
