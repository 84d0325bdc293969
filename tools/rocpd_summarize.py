"""Summarize rocprofv3 rocpd sqlite output per kernel.

Usage: python tools/rocpd_summarize.py OUT.json DB [DB...]
For each results .db: per-kernel launch count, total/avg duration from
rocpd_kernel_dispatch, and per-kernel PMC counter sums (FETCH_SIZE /
WRITE_SIZE runs) when rocpd_pmc_event is populated. Column names are
resolved by inspection so minor schema drift across ROCm versions doesn't
break the extraction; on failure the discovered schema is embedded in the
output for debugging.
"""

import glob
import json
import sqlite3
import sys


def cols(con, table):
    try:
        return [r[1] for r in con.execute(f"PRAGMA table_info({table})")]
    except sqlite3.Error:
        return []


def pick(names, *subs):
    for n in names:
        ln = n.lower()
        if all(s in ln for s in subs):
            return n
    return None


def tables(con):
    return [r[0] for r in con.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]


def summarize_db(path, out):
    con = sqlite3.connect(path)
    tabs = tables(con)
    dbg = {t: cols(con, t) for t in tabs}

    dis_t = pick(tabs, "kernel_dispatch") or pick(tabs, "dispatch")
    sym_t = pick(tabs, "kernel_symbol") or pick(tabs, "kernel_info") \
        or pick(tabs, "info_kernel")
    if not dis_t:
        out.setdefault("_schema_debug", {})[path] = dbg
        return

    dc = cols(con, dis_t)
    start = pick(dc, "start")
    end = pick(dc, "end")
    kid = pick(dc, "kernel", "id") or pick(dc, "symbol", "id")
    name_expr = None
    join = ""
    if sym_t and kid:
        sc = cols(con, sym_t)
        sid = pick(sc, "id") or sc[0]
        sname = pick(sc, "display") or pick(sc, "formatted") \
            or pick(sc, "kernel", "name") or pick(sc, "name")
        if sname:
            name_expr = f"s.{sname}"
            alt = pick(sc, "kernel", "name")
            if alt and alt != sname:
                name_expr = f"COALESCE(NULLIF(s.{sname}, ''), s.{alt})"
            join = f"JOIN {sym_t} s ON d.{kid} = s.{sid}"
    if name_expr is None:
        nm = pick(dc, "name")
        if nm:
            name_expr = f"d.{nm}"
    if not (start and end and name_expr):
        out.setdefault("_schema_debug", {})[path] = dbg
        return

    q = (f"SELECT {name_expr} AS name, COUNT(*), SUM(d.{end}-d.{start}), "
         f"AVG(d.{end}-d.{start}) FROM {dis_t} d {join} GROUP BY 1")
    for name, n, total, avg in con.execute(q):
        short = shorten(name)
        k = out.setdefault("kernels", {}).setdefault(
            short, {"launches": 0, "total_ns": 0})
        k["launches"] += n
        k["total_ns"] += int(total or 0)
        k["avg_ns"] = k["total_ns"] // max(k["launches"], 1)

    # PMC events, if present
    pmc_t = pick(tabs, "pmc_event")
    pmc_i = pick(tabs, "info_pmc") or pick(tabs, "pmc_info")
    if pmc_t:
        pc = cols(con, pmc_t)
        val = pick(pc, "value")
        # event links either to dispatch id or its own correlation
        link = pick(pc, "dispatch") or pick(pc, "event", "id") \
            or pick(pc, "corr")
        pid = pick(pc, "pmc", "id")
        cname_expr, cjoin = None, ""
        if pmc_i and pid:
            ic = cols(con, pmc_i)
            iid = pick(ic, "id") or ic[0]
            iname = pick(ic, "name") or pick(ic, "symbol")
            if iname:
                cname_expr = f"i.{iname}"
                cjoin = f"JOIN {pmc_i} i ON p.{pid} = i.{iid}"
        if val and link and cname_expr:
            dl = link if link in dc else (pick(dc, "id") or dc[0])
            q = (f"SELECT {name_expr}, {cname_expr}, SUM(p.{val}), COUNT(*) "
                 f"FROM {pmc_t} p JOIN {dis_t} d ON p.{link} = d.{dl} {join} "
                 f"{cjoin} GROUP BY 1, 2")
            try:
                rows = list(con.execute(q))
                for name, counter, total, n in rows:
                    short = shorten(name)
                    k = out.setdefault("kernels", {}).setdefault(
                        short, {"launches": 0, "total_ns": 0})
                    c = k.setdefault("counters", {})
                    c[str(counter)] = c.get(str(counter), 0) + float(total or 0)
                    c[f"{counter}_dispatches"] = \
                        c.get(f"{counter}_dispatches", 0) + n
                if not rows:
                    npmc = next(con.execute(f"SELECT COUNT(*) FROM {pmc_t}"))[0]
                    if npmc:
                        out.setdefault("_schema_debug", {})[path + ":pmc"] = {
                            "note": f"join matched 0 of {npmc} pmc rows",
                            "query": q,
                            "pmc_sample": [list(r) for r in con.execute(
                                f"SELECT * FROM {pmc_t} LIMIT 3")],
                            "dispatch_sample": [list(r) for r in con.execute(
                                f"SELECT {', '.join(dc[:8])} FROM {dis_t} "
                                f"LIMIT 3")],
                            "dispatch_cols": dc,
                            "pmc_cols": pc,
                        }
            except sqlite3.Error as exc:
                out.setdefault("_schema_debug", {})[path + ":pmc"] = \
                    {"error": str(exc), **dbg}
        elif any(con.execute(f"SELECT 1 FROM {pmc_t} LIMIT 1")):
            out.setdefault("_schema_debug", {})[path + ":pmc"] = dbg
    con.close()


def shorten(name):
    """'void (anonymous namespace)::k_phase_s(args)' -> 'k_phase_s';
    'void rocprim::...::trampoline_kernel<...>(args)' -> 'trampoline_kernel'."""
    s = str(name).replace("(anonymous namespace)::", "")
    head = s.split("(")[0].split("<")[0].strip()
    parts = head.split()
    return parts[-1] if parts else s[:60]


def main():
    out_path = sys.argv[1]
    out = {}
    for arg in sys.argv[2:]:
        for path in sorted(glob.glob(arg, recursive=True)):
            summarize_db(path, out)
    if "kernels" in out:
        out["kernels"] = dict(sorted(
            out["kernels"].items(), key=lambda kv: -kv[1]["total_ns"]))
    with open(out_path, "w") as fh:
        json.dump(out, fh, indent=1)
    for name, k in list(out.get("kernels", {}).items())[:10]:
        print(f"{name}: {k['launches']}x avg {k.get('avg_ns', 0)/1e3:.1f}us "
              f"total {k['total_ns']/1e6:.1f}ms "
              + " ".join(f"{c}={v:.3e}" for c, v in k.get("counters", {}).items()
                         if not c.endswith("_dispatches")))
    if "_schema_debug" in out:
        print("schema debug entries:", list(out["_schema_debug"]))


if __name__ == "__main__":
    main()
