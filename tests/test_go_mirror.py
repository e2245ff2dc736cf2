"""Go mirror drift guard.

No Go toolchain exists in this image, so go/nezha/nezha.go can never be
compile-verified here. This guard keeps it from rotting anyway: it parses
the C ABI declarations in include/nezha.h and the `C.nz_*` call sites in
the Go source, and fails if the mirror calls a symbol that does not exist
or passes a wrong argument count.
"""
import os
import re

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
HEADER = os.path.join(REPO, "include", "nezha.h")
GO = os.path.join(REPO, "go", "nezha", "nezha.go")


def header_decls():
    """nz_* function name -> parameter count, from include/nezha.h."""
    src = open(HEADER).read()
    src = re.sub(r"/\*.*?\*/", " ", src, flags=re.S)  # strip comments
    src = re.sub(r"//[^\n]*", " ", src)
    decls = {}
    # declarations may span lines; match up to the closing ');'
    for m in re.finditer(r"\b(nz_\w+)\s*\(([^;]*?)\)\s*;", src, flags=re.S):
        name, params = m.group(1), m.group(2).strip()
        if params in ("", "void"):
            decls[name] = 0
        else:
            depth = 0
            count = 1
            for ch in params:
                if ch == "(":
                    depth += 1
                elif ch == ")":
                    depth -= 1
                elif ch == "," and depth == 0:
                    count += 1
            decls[name] = count
    return decls


def go_calls():
    """(symbol, argc) for every C.nz_* call in the Go mirror."""
    src = open(GO).read()
    src = re.sub(r"//[^\n]*", " ", src)
    calls = []
    for m in re.finditer(r"\bC\.(nz_\w+)\(", src):
        name = m.group(1)
        # count top-level commas in the balanced paren span
        i = m.end()
        depth = 1
        argc = 0
        saw_arg = False
        while depth > 0 and i < len(src):
            ch = src[i]
            if ch in "([":
                depth += 1
            elif ch in ")]":
                depth -= 1
            elif ch == "," and depth == 1:
                argc += 1
            elif not ch.isspace() and depth >= 1:
                saw_arg = True
            i += 1
        calls.append((name, argc + 1 if saw_arg else 0))
    return calls


def test_go_mirror_symbols_exist_with_matching_arity():
    decls = header_decls()
    assert len(decls) >= 60, f"header parse broke? found {len(decls)} decls"
    calls = go_calls()
    assert len(calls) >= 30, f"Go parse broke? found {len(calls)} calls"
    errors = []
    for name, argc in calls:
        if name not in decls:
            errors.append(f"{name}: called from Go but absent from nezha.h")
        elif decls[name] != argc:
            errors.append(f"{name}: Go passes {argc} args, header declares "
                          f"{decls[name]}")
    assert not errors, "\n".join(errors)


def test_go_mirror_covers_core_api():
    """The mirror must keep binding the core trainer/checkpoint surface."""
    src = open(GO).read()
    for sym in ["nz_ctx_new", "nz_tensor_new", "nz_trainer_new_mlp",
                "nz_trainer_step", "nz_ckpt_save", "nz_ckpt_load",
                "nz_comm_init_rccl", "nz_trainer_attach_ddp"]:
        assert f"C.{sym}(" in src, f"Go mirror no longer binds {sym}"


def test_pingpong_schedule_simulator():
    """The GEMM ping-pong stage schedule must pass its FIFO/certificate
    simulator (tools/pp_schedule_sim.py) — this is the guard that caught
    the r2 deep-schedule race before it could ship."""
    import subprocess
    import sys as _sys
    r = subprocess.run([_sys.executable,
                        os.path.join(REPO, "tools", "pp_schedule_sim.py")],
                       capture_output=True, timeout=60)
    assert r.returncode == 0, r.stdout.decode() + r.stderr.decode()


def test_pp64q4_schedule_simulator():
    """The 4-phase pp64q variant's stage schedule (triple-buffered A1/B1,
    vmcnt<=4 at end-P0/P2) must pass its own verifier
    (tools/pp64q4_sim.py). The kernel is an opt-in (NEZHA_GEMM_64Q=4),
    but the simulator keeps its documented schedule honest."""
    import subprocess
    import sys as _sys
    r = subprocess.run([_sys.executable,
                        os.path.join(REPO, "tools", "pp64q4_sim.py")],
                       capture_output=True, timeout=60)
    assert r.returncode == 0, r.stdout.decode() + r.stderr.decode()
