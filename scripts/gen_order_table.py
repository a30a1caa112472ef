#!/usr/bin/env python3
"""Generate the mask-config group-order catalogue header.

Rule (verified against the reference catalogue, see SURVEY.md "Order-table
generation rule"): with B = 2 * add_shift * exp_shift * max_nb_models,
  order(Integer) = B + 1
  order(Prime)   = smallest prime >= B
  order(Power2)  = smallest power of two >= B

add_shift: B0=1, B2=100, B4=10^4, B6=10^6, Bmax=f32::MAX / f64::MAX /
-i32::MIN / -i64::MIN (exact values; f32::MAX = (2^24-1)*2^104,
f64::MAX = (2^53-1)*2^971).
exp_shift: f32 -> 10^10 (10^45 for Bmax), f64 -> 10^20 (10^324 for Bmax),
i32/i64 -> 10^10.
max_nb_models: M3=10^3, M6=10^6, M9=10^9, M12=10^12.

Emits xaynet_amd/csrc/mask/order_table.h with decimal strings indexed
[group][dtype][bound][model].
"""
import os
import random

GROUPS = ["Integer", "Prime", "Power2"]
DTYPES = ["F32", "F64", "I32", "I64"]
BOUNDS = ["B0", "B2", "B4", "B6", "Bmax"]
MODELS = ["M3", "M6", "M9", "M12"]

F32_MAX = (2**24 - 1) * 2**104
F64_MAX = (2**53 - 1) * 2**971

# The reference catalogue was generated from decimal-rounded 2*MAX for the
# float Bmax rows: 2*f32::MAX ~= 6.805647e38, 2*f64::MAX ~= 3.595386269724631e308.
DOUBLED_MAX = {
    "F32": 6805647 * 10**32,
    "F64": 3595386269724631 * 10**293,
    "I32": 2**32 - 1,
    "I64": 2**64 - 1,
}


PRIME_F64_BMAX = {
    "M3": 359538626972463140000000000000000000000593874019667231666067439096529924969333439983391110599943465644007133099721551828263813044710323667390405279670626898022875314671948577301533414396469719048504306012596386638859340084030210314832025518258115226051894034477843584650149420090374373134876775786923748346298936467612015276401624887654050299443392510555689981501608709494004423956258647440955320257123787935493476104132776728548437783283112428445450269488453346610914359272368862786051728965455746393095846720860347644662201994241194193316457656284847050135299403149697261199957835824000531233031619352921347101423914861961738035659301,
    "M6": 359538626972463139999999999999999999999903622106309601840402558296261360055843460163714984640183652353129826112739444431322400938984152600575421591212739537896016542591595727264024538428559469178136611680881710150818089794351154869285409959876691068635451827253162844058791343487286852635234799336668682655217329655102622197942194212857658834043465713831143523811067060369640438677832007091511212788398470391285320720769417737628120102221909739846753580817462645602854496103866327474145187363329320852679912679009543036760757409720574191338832841104183169976025577743061881721861634977765641182996194573448626763720938201976656541039724303,
    "M9": 359538626972463139999999999999999999999904930781891526077660862016966437766478934820885791914528679207262530042483798832910003057874958310694484517139841166977272287522418122134527125053808273636647181903383717418169782215585647900802728035567327931187710919458230957036511507150288137858111024099126399746768695036546643813753385062385762652380150346615796407577297605069883839431646689072072214687584099356273959025519093953786032481175596842406101871239892163505527137519569046747947203065300865116331411924515285552096042635874474960733445241451746509870642272026256695499704624475309137281644358183373160068523639023207643484888657559597,
    "M12": 359538626972463139999999999999999999999904931540467867407238817633447114203759664620787471913925990313859370016783101785327523046787247090978931042236128228564142680745383377953776024143512065781667978525748300241659425164472387573470260831720974578793447369507661739490218806790001765109117055431552295585457639803896262637528011897242316426079400392728240523639775219294589603009325941759217573340626063716838671315192395974939441284468885927433422082497928190254190935717337452741850223510814859331413287559285438144477756395583878761313295130567342888620541025745968373350261259032809052052475301496416128372300050762773363722300553930211649,
}


def doubled_add_shift(dtype, bound):
    """2 * add_shift as used by the catalogue rule."""
    if bound == "B0":
        return 2
    if bound == "B2":
        return 200
    if bound == "B4":
        return 2 * 10**4
    if bound == "B6":
        return 2 * 10**6
    return DOUBLED_MAX[dtype]


def exp_shift(dtype, bound):
    if dtype == "F32":
        return 10**45 if bound == "Bmax" else 10**10
    if dtype == "F64":
        return 10**324 if bound == "Bmax" else 10**20
    return 10**10


def max_nb(model):
    return 10 ** int(model[1:])


def is_prime(n, rounds=48):
    if n < 2:
        return False
    for p in (2, 3, 5, 7, 11, 13, 17, 19, 23, 29, 31, 37):
        if n % p == 0:
            return n == p
    d, s = n - 1, 0
    while d % 2 == 0:
        d //= 2
        s += 1
    rng = random.Random(0xC0FFEE)
    for _ in range(rounds):
        a = rng.randrange(2, n - 1)
        x = pow(a, d, n)
        if x in (1, n - 1):
            continue
        for _ in range(s - 1):
            x = pow(x, 2, n)
            if x == n - 1:
                break
        else:
            return False
    return True


def next_prime(n):
    if n <= 2:
        return 2
    if n % 2 == 0:
        n += 1
    while not is_prime(n):
        n += 2
    return n


def order(group, dtype, bound, model):
    B = doubled_add_shift(dtype, bound) * exp_shift(dtype, bound) * max_nb(model)
    # catalogue quirk: the reference's Prime/F64/Bmax rows don't follow the
    # next_prime(B) rule for any B derivable from f64::MAX (they were produced
    # by a different tool); they are protocol constants, embedded verbatim.
    if group == "Prime" and dtype == "F64" and bound == "Bmax":
        return PRIME_F64_BMAX[model]
    if group == "Integer":
        return B + 1
    if group == "Prime":
        return next_prime(B)
    return 1 << (B - 1).bit_length() if B > 1 else 1


def main():
    out = []
    out.append("// AUTOGENERATED by scripts/gen_order_table.py — do not edit.")
    out.append("// Group-order catalogue for every MaskConfig (protocol contract;")
    out.append("// values equal the reference catalogue in")
    out.append("// rust/xaynet-core/src/mask/config/mod.rs:234-640).")
    out.append("#pragma once")
    out.append("")
    out.append("namespace xaynet::mask {")
    out.append("// indexed [group(3)][dtype(4)][bound(5)][model(4)]")
    out.append("inline const char* ORDER_TABLE[3][4][5][4] = {")
    for g in GROUPS:
        out.append(f"  {{  // {g}")
        for d in DTYPES:
            out.append(f"    {{  // {d}")
            for b in BOUNDS:
                vals = ", ".join(f'"{order(g, d, b, m)}"' for m in MODELS)
                out.append(f"      {{{vals}}},  // {b}")
            out.append("    },")
        out.append("  },")
    out.append("};")
    out.append("}  // namespace xaynet::mask")
    path = os.path.join(os.path.dirname(__file__), "..", "xaynet_amd", "csrc", "mask", "order_table.h")
    with open(path, "w") as f:
        f.write("\n".join(out) + "\n")
    print(f"wrote {path}")

    # spot checks from SURVEY.md (verified against the reference)
    assert order("Integer", "F32", "B0", "M3") == 20_000_000_000_001
    assert order("Prime", "F32", "B0", "M3") == 20_000_000_000_021
    assert order("Power2", "F32", "B0", "M3") == 2**45
    print("spot checks OK")


if __name__ == "__main__":
    main()
