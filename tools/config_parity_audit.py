"""Config-parity audit: compare our composed algo config defaults against the
reference's (defaults-chain merged, ${algo.*} interpolation resolved on both
sides).  Run with the reference mounted at /root/reference:

    python tools/config_parity_audit.py [--all]

False-positive classes that are filtered by default: hydra instantiation keys
(_target_ etc., we use string names), activation-class spellings, float
formatting, keys we deliberately default differently for the offline image
(mlp/cnn key lists point at the in-repo envs).
"""

import glob
import os
import re
import sys

import yaml

REF = "/root/reference/sheeprl/configs"
OURS = os.path.join(os.path.dirname(__file__), "..", "sheeprl_amd", "configs")
ACT = {"torch.nn.SiLU": "silu", "torch.nn.ReLU": "relu", "torch.nn.Tanh": "tanh", "torch.nn.ELU": "elu"}
# sac_ae: our critic.target_network_frequency=2 implements the reference's
# per_rank_target_network_update_freq=2; its inherited target_network_frequency=1
# is unused there.
IGN = re.compile(
    r"(_target_|_convert_|_partial_|layer_norm_cls|layer_norm_kw|\.cls$|^name$|cnn_keys|mlp_keys"
    r"|target_network_frequency)"
)


def deep_merge(a, b):
    out = dict(a)
    for k, v in b.items():
        if isinstance(v, dict) and isinstance(out.get(k), dict):
            out[k] = deep_merge(out[k], v)
        else:
            out[k] = v
    return out


def resolve_file(root, group, name):
    raw = yaml.safe_load(open(os.path.join(root, group, name + ".yaml"))) or {}
    out = {}
    for d in raw.get("defaults", []):
        if d == "_self_" or isinstance(d, dict):
            continue
        out = deep_merge(out, resolve_file(root, group, d))
    raw.pop("defaults", None)
    return deep_merge(out, raw)


def interp(tree):
    def get(path):
        cur = tree
        for part in path.split("."):
            if not isinstance(cur, dict) or part not in cur:
                return None
            cur = cur[part]
        return cur

    def walk(d):
        for k, v in list(d.items()):
            if isinstance(v, dict):
                walk(v)
            elif isinstance(v, str):
                m = re.fullmatch(r"\$\{algo\.([\w.]+)\}", v)
                if m and get(m.group(1)) is not None:
                    d[k] = get(m.group(1))

    for _ in range(4):
        walk(tree)
    return tree


def flat(d, pre=""):
    out = {}
    for k, v in (d or {}).items():
        if isinstance(v, dict):
            out.update(flat(v, pre + k + "."))
        else:
            out[pre + k] = v
    return out


def norm(v):
    if isinstance(v, str) and v in ACT:
        return ACT[v]
    if isinstance(v, bool):
        return v
    if isinstance(v, (int, float)):
        return float(v)
    if isinstance(v, str):
        # yaml 1.1 parses "3e-4" as a string; normalize numeric spellings
        try:
            return float(v)
        except ValueError:
            return v
    return v


def main():
    show_absent = "--all" in sys.argv
    n = 0
    for ref_f in sorted(glob.glob(REF + "/algo/*.yaml")):
        name = os.path.splitext(os.path.basename(ref_f))[0]
        our_f = os.path.join(OURS, "algo", name + ".yaml")
        if not os.path.exists(our_f):
            print(f"MISSING: algo/{name}")
            n += 1
            continue
        r = flat(interp(resolve_file(REF, "algo", name)))
        o = flat(interp(resolve_file(OURS, "algo", name)))
        for k in sorted(r):
            if IGN.search(k) or r[k] == "???":
                continue
            if k not in o:
                if show_absent:
                    print(f"{name}: {k} ref={r[k]} ours=<absent>")
                continue
            rv, ov = norm(r[k]), norm(o[k])
            if isinstance(rv, float) and isinstance(ov, float):
                if abs(rv - ov) <= 1e-12 * max(1.0, abs(rv)):
                    continue
            elif rv == ov:
                continue
            if re.search(r"(dense_act|cnn_act|activation)", k) and isinstance(ov, str):
                if isinstance(rv, str) and ACT.get(rv, rv).lower() == ov.lower():
                    continue
            print(f"{name}: {k} ref={r[k]} ours={o[k]}")
            n += 1
    print("value diffs:", n)


if __name__ == "__main__":
    main()
