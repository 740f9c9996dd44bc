#!/usr/bin/env python3
"""Static ISA evidence for the gfx950 kernels (runs with no GPU).

Compiles every .hip source device-only and disassembles the code objects
with llvm-objdump, then reports per-kernel instruction-mix facts that the
performance claims rest on:

  * vectorized 16 B/lane feature traffic  -> global_load/store_dwordx4
  * MFMA matrix-core usage in gather_mm   -> v_mfma_f32_16x16x4* ops
  * LDS staging / reductions              -> ds_read/ds_write ops
  * wave-level reductions (shfl_xor)      -> ds_swizzle / v_permlane
  * fp32 atomics for scatter backward     -> global_atomic_add_f32
  * spill check                           -> scratch_* absent

Writes profiles/isa_gfx950.md. Complements the occupancy table in
kernel_resources_gfx950.txt; the dynamic (rocprofv3) numbers remain
blocked on GPU availability (see README.md).
"""
import collections
import os
import re
import subprocess
import sys
import tempfile

HERE = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.dirname(HERE)
sys.path.insert(0, REPO)

CLASSES = [
    ("global_load_dwordx4", r"global_load_dwordx4"),
    ("global_load_dwordx2", r"global_load_dwordx2"),
    ("global_load_dword\b", r"global_load_dword(?!x)"),
    ("global_store_dwordx4", r"global_store_dwordx4"),
    ("global_atomic_add_f32", r"global_atomic_add_f32"),
    ("ds_read", r"\bds_read"),
    ("ds_write", r"\bds_write"),
    ("ds_add_f32 (LDS atomic)", r"\bds_add(_rtn)?_f32"),
    ("ds_bpermute/swizzle (shfl)", r"ds_bpermute|ds_swizzle|v_permlane"),
    ("v_mfma", r"\bv_mfma"),
    ("scratch (spill!)", r"\bscratch_"),
]

KERNELS_OF_INTEREST = [
    "spmm_kernel", "spmm_long_kernel", "spmm_scatter_kernel",
    "sddmm_dot_kernel", "edge_softmax_fwd_kernel",
    "edge_softmax_fwd_long_kernel", "edge_softmax_bwd_long_kernel",
    "segment_reduce_kernel", "gather_rows_kernel", "gat_score_fwd_kernel",
    "sample_kernel", "compact_claim_kernel", "pack_padded_kernel",
    "gather_mm_kernel", "adagrad", "pdist", "cpdist",
]


def compile_device_only(src, out, extra_flags):
    cmd = ["hipcc", "-c", "--offload-device-only",
           "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC",
           "-munsafe-fp-atomics", "-D__HIP_PLATFORM_AMD__=1",
           "-DUSE_ROCM=1", "-D_GLIBCXX_USE_CXX11_ABI=1", "-x", "hip",
           src, "-o", out] + extra_flags
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        raise RuntimeError(f"{src}: {r.stderr[-2000:]}")


def disassemble(obj):
    # --offload-device-only still emits a clang offload bundle; peel the
    # gfx950 ELF out before disassembling
    hsaco = obj + ".hsaco"
    r = subprocess.run(
        ["/opt/rocm/lib/llvm/bin/clang-offload-bundler", "--unbundle",
         "--type=o", "--targets=hipv4-amdgcn-amd-amdhsa--gfx950",
         f"--input={obj}", f"--output={hsaco}"],
        capture_output=True, text=True)
    assert r.returncode == 0, r.stderr[-500:]
    r = subprocess.run(
        ["/opt/rocm/lib/llvm/bin/llvm-objdump", "-d", hsaco],
        capture_output=True, text=True)
    assert r.returncode == 0, r.stderr[-500:]
    return r.stdout


def split_kernels(disasm):
    """symbol -> list of instruction lines."""
    out = {}
    cur = None
    for line in disasm.splitlines():
        m = re.match(r"^[0-9a-f]+ <(.+)>:$", line)
        if m:
            cur = m.group(1)
            out[cur] = []
        elif cur is not None and "\t" in line:
            out[cur].append(line)
    return out


def main():
    from torch.utils import cpp_extension as ce

    import sysconfig

    includes = ce.include_paths() + [sysconfig.get_paths()["include"],
                                     os.path.join(REPO, "dgl_operator_amd",
                                                  "csrc")]
    flags = [f"-I{p}" for p in includes]
    srcs = ["gnn_ops.hip", "sampling.hip", "adagrad.hip", "kge.hip",
            "gather_mm.hip"]
    rows = []
    facts = collections.defaultdict(dict)
    with tempfile.TemporaryDirectory() as td:
        for s in srcs:
            src = os.path.join(REPO, "dgl_operator_amd", "csrc", s)
            obj = os.path.join(td, s + ".o")
            print(f"[isa] compiling {s} (device-only)...", flush=True)
            compile_device_only(src, obj, flags)
            kernels = split_kernels(disassemble(obj))
            for name, instrs in kernels.items():
                short = name.split("(")[0]
                if not any(k in short for k in KERNELS_OF_INTEREST):
                    continue
                text = "\n".join(instrs)
                counts = {label: len(re.findall(pat, text))
                          for label, pat in CLASSES}
                counts["total_instrs"] = len(instrs)
                facts[s][short] = counts

    lines = [
        "# Static ISA evidence (gfx950 code objects, llvm-objdump)",
        "",
        "Per-kernel instruction-mix counts backing the design claims.",
        "Dynamic counters (rocprofv3) remain blocked on GPU availability;",
        "see README.md. Regenerate: `python profiles/isa_check.py`.",
        "",
    ]
    claims = []
    for s, kmap in sorted(facts.items()):
        lines.append(f"## {s}")
        lines.append("")
        header = ["kernel"] + [c[0] for c in CLASSES] + ["total"]
        lines.append("| " + " | ".join(header) + " |")
        lines.append("|" + "---|" * len(header))
        for kname, counts in sorted(kmap.items()):
            row = [f"`{kname[:58]}`"] + [
                str(counts[c[0]]) for c in CLASSES
            ] + [str(counts["total_instrs"])]
            lines.append("| " + " | ".join(row) + " |")
        lines.append("")

    # headline claims, checked mechanically
    def find(sub, src=None):
        for s, kmap in facts.items():
            if src and s != src:
                continue
            for k, c in kmap.items():
                if sub in k:
                    yield s, k, c

    mfma = sum(c["v_mfma"] for _, _, c in find("gather_mm"))
    claims.append(f"- gather_mm contains {mfma} v_mfma instructions "
                  f"({'OK' if mfma > 0 else 'MISSING — NOT on matrix cores'})"
                  ": the fused gather+projection runs on MFMA.")
    vec = sum(c["global_load_dwordx4"] for _, _, c in find("spmm_kernel"))
    claims.append(f"- spmm_kernel VEC=4 instantiations issue "
                  f"{vec} global_load_dwordx4 (16 B/lane vectorized "
                  f"feature gathers) — {'OK' if vec > 0 else 'MISSING'}.")
    lds = sum(c["ds_add_f32 (LDS atomic)"] + c["ds_read"] + c["ds_write"]
              for _, _, c in find("spmm_long_kernel"))
    claims.append(f"- spmm_long_kernel uses {lds} LDS ops (block-parallel "
                  f"hub-row reduction through shared memory).")
    at = sum(c["global_atomic_add_f32"] for _, _, c in find("scatter"))
    claims.append(f"- spmm_scatter backward uses {at} global_atomic_add_f32 "
                  f"(fp32 scatter accumulation, -munsafe-fp-atomics).")
    sw = sum(c["ds_bpermute/swizzle (shfl)"] for _, _, c in
             find("edge_softmax_fwd_long"))
    claims.append(f"- edge_softmax long-row kernels lower __shfl_xor to "
                  f"{sw} ds_bpermute/swizzle ops (wave-level online-softmax "
                  f"merge).")
    spills = [(s, k) for s, k, c in find("") if c["scratch (spill!)"] > 0]
    claims.append(
        "- scratch (spill) instructions: "
        + ("NONE in any inspected kernel." if not spills
           else f"PRESENT in {spills}!"))
    lines += ["## Mechanically-checked claims", ""] + claims + [""]

    out = os.path.join(HERE, "isa_gfx950.md")
    with open(out, "w") as f:
        f.write("\n".join(lines))
    print("\n".join(claims))
    print(f"wrote {out}")


if __name__ == "__main__":
    main()
