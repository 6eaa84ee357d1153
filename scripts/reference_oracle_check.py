#!/usr/bin/env python3
"""Bit-level parity check against the reference's OWN compiled kernels.

Builds (in /tmp, from the read-only reference mount -- nothing is copied
into this repo) two oracles:

1. ``ref_ops``: the reference's actual `op_krum/cpu.cpp` and
   `op_bulyan/cpu.cpp` kernels, compiled against a minimal stub of the
   TensorFlow kernel API (the kernels only use Tensor::flat and
   allocate_temp) plus the reference's own threadpool.
2. ``libdeprecated.so``: the reference's ctypes GAR library
   (`aggregators/deprecated_native/native.cpp`) exposing squared_distance /
   median / averaged_median / average_nan.

Then runs a matrix of inputs (normal, NaN-laced, adversarial outliers, tie
cases) through BOTH the reference binaries and this framework's
implementations (`aggregathor_amd.ops.reference`), asserting agreement --
exact for element-selection rules, ~1-2 ulp for fp32 reductions.

Usage: python scripts/reference_oracle_check.py  (needs /root/reference
and g++; exits non-zero on any mismatch). The committed parity results
live in results/reference_parity.md.
"""

import ctypes
import os
import pathlib
import struct
import subprocess
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

REF = pathlib.Path("/root/reference")
WORK = pathlib.Path("/tmp/refcheck")

STUB_OP_KERNEL = r"""
// Minimal stand-in for the TensorFlow kernel API surface used by the
// reference's op_krum/op_bulyan CPU kernels (verification harness only).
#pragma once
#include <cstdint>
#include <cstddef>
#include <initializer_list>
#include <memory>
namespace Eigen { struct ThreadPoolDevice {}; struct GpuDevice {}; }
namespace tensorflow {
using uint32 = std::uint32_t;
using uint64 = std::uint64_t;
enum DataType { DT_FLOAT = 1, DT_DOUBLE = 2 };
template<class T> struct DataTypeToEnum;
template<> struct DataTypeToEnum<float>  { static constexpr DataType value = DT_FLOAT; };
template<> struct DataTypeToEnum<double> { static constexpr DataType value = DT_DOUBLE; };
struct TensorShape {
  long long n;
  TensorShape(std::initializer_list<long long> l) : n(*l.begin()) {}
};
struct Status { bool ok() const { return true; } };
class Tensor {
 public:
  Tensor() = default;
  explicit Tensor(std::size_t nelems)
      : buf_(new double[nelems], std::default_delete<double[]>()),
        elems_(nelems) {}
  template<class T> struct Flat {
    T* p;
    T* data() const { return p; }
  };
  template<class T> Flat<T> flat() const {
    return Flat<T>{reinterpret_cast<T*>(buf_.get())};
  }
  std::shared_ptr<double> buf_;
  std::size_t elems_ = 0;
};
class OpKernelContext {
 public:
  Status allocate_temp(DataType, TensorShape shape, Tensor* out) {
    *out = Tensor(static_cast<std::size_t>(shape.n));
    return Status{};
  }
};
}  // namespace tensorflow
#define OP_REQUIRES_OK(ctx, expr) do { (void)(expr); } while (0)
"""

STUB_SHAPE_INF = "#pragma once\nnamespace tensorflow { namespace shape_inference { } }\n"

HARNESS_MAIN = r"""
#include <cstdint>
#include <cstdio>
#include <tensorflow/core/framework/op_kernel.h>
#include <common.hpp>
using namespace tensorflow;
namespace Krum {
template<class Device, class T> class Kernel: public Static {
 public:
  static void process(OpKernelContext&, size_t const, size_t const,
                      size_t const, size_t const, Tensor const&, Tensor&);
};
}
namespace Bulyan {
template<class Device, class T> class Kernel: public Static {
 public:
  static void process(OpKernelContext&, size_t const, size_t const,
                      size_t const, size_t const, Tensor const&, Tensor&);
};
}
using CPUDev = Eigen::ThreadPoolDevice;
int main(int argc, char** argv) {
  if (argc != 3) return 2;
  std::FILE* f = std::fopen(argv[1], "rb");
  if (!f) return 3;
  std::uint64_t hdr[5];  // kind(0=krum,1=bulyan), n, f, m, d
  if (std::fread(hdr, sizeof hdr, 1, f) != 1) return 4;
  std::uint64_t kind = hdr[0], n = hdr[1], ff = hdr[2], m = hdr[3], d = hdr[4];
  Tensor input(n * d), output(d);
  if (std::fread(input.flat<float>().data(), sizeof(float), n * d, f) != n * d)
    return 5;
  std::fclose(f);
  OpKernelContext ctx;
  if (kind == 0)
    Krum::Kernel<CPUDev, float>::process(ctx, n, ff, d, m, input, output);
  else
    Bulyan::Kernel<CPUDev, float>::process(ctx, n, ff, d, m, input, output);
  std::FILE* o = std::fopen(argv[2], "wb");
  if (!o) return 6;
  std::fwrite(output.flat<float>().data(), sizeof(float), d, o);
  std::fclose(o);
  return 0;
}
"""


def build_oracles():
    """Compile the reference kernels in /tmp; returns (ref_ops, lib) paths."""
    assert REF.exists(), "reference mount not available"
    inc = WORK / "tensorflow" / "core" / "framework"
    inc.mkdir(parents=True, exist_ok=True)
    (inc / "op_kernel.h").write_text(STUB_OP_KERNEL)
    (inc / "shape_inference.h").write_text(STUB_SHAPE_INF)
    (WORK / "main.cpp").write_text(HARNESS_MAIN)

    def run(cmd):
        subprocess.run(cmd, check=True, cwd=str(WORK))

    native_inc = str(REF / "native" / "include")
    run(["g++", "-std=c++14", "-O2", "-c", "-I.", f"-I{native_inc}",
         f"-I{REF}/native/op_krum", str(REF / "native/op_krum/cpu.cpp"),
         "-o", "krum.o"])
    run(["g++", "-std=c++14", "-O2", "-c", "-include", "cstdlib", "-I.",
         f"-I{native_inc}", f"-I{REF}/native/op_bulyan",
         str(REF / "native/op_bulyan/cpu.cpp"), "-o", "bulyan.o"])
    run(["g++", "-std=c++14", "-O2", "-c", f"-I{native_inc}",
         str(REF / "native/so_threadpool/threadpool.cpp"), "-o", "pool.o"])
    run(["g++", "-std=c++14", "-O2", "-include", "cstdlib", "-I.",
         f"-I{native_inc}", "main.cpp", "krum.o", "bulyan.o", "pool.o",
         "-o", "ref_ops", "-lpthread"])
    run(["g++", "-std=c++14", "-O2", "-shared", "-fPIC",
         str(REF / "aggregators/deprecated_native/native.cpp"),
         "-o", "libdeprecated.so", "-lpthread"])
    return WORK / "ref_ops", WORK / "libdeprecated.so"


def ref_op(binary, kind, g, f, m):
    n, d = g.shape
    inp = WORK / "in.bin"
    out = WORK / "out.bin"
    with open(inp, "wb") as fo:
        fo.write(struct.pack("<5Q", kind, n, f, m, d))
        fo.write(np.ascontiguousarray(g, dtype=np.float32).tobytes())
    subprocess.run([str(binary), str(inp), str(out)], check=True)
    return np.fromfile(out, dtype=np.float32)


def load_lib(path):
    lib = ctypes.CDLL(str(path))
    lib.squared_distance_float.restype = ctypes.c_float
    return lib


def _fp(a):
    return a.ctypes.data_as(ctypes.POINTER(ctypes.c_float))


def lib_median(lib, g):
    g = np.ascontiguousarray(g, dtype=np.float32).copy()  # mangled by callee
    n, d = g.shape
    out = np.empty(d, dtype=np.float32)
    lib.median_float(ctypes.c_size_t(d), ctypes.c_size_t(n), _fp(g), _fp(out))
    return out


def lib_averaged_median(lib, g, beta):
    g = np.ascontiguousarray(g, dtype=np.float32).copy()
    n, d = g.shape
    out = np.empty(d, dtype=np.float32)
    lib.averaged_median_float(ctypes.c_size_t(d), ctypes.c_size_t(n),
                              ctypes.c_size_t(beta), _fp(g), _fp(out))
    return out


def lib_average_nan(lib, g):
    g = np.ascontiguousarray(g, dtype=np.float32).copy()
    n, d = g.shape
    out = np.empty(d, dtype=np.float32)
    lib.average_nan_float(ctypes.c_size_t(d), ctypes.c_size_t(n), _fp(g),
                          _fp(out))
    return out


def lib_sqdist(lib, a, b):
    a = np.ascontiguousarray(a, dtype=np.float32)
    b = np.ascontiguousarray(b, dtype=np.float32)
    return lib.squared_distance_float(ctypes.c_size_t(a.size), _fp(a), _fp(b))


def make_cases():
    """(name, g, f) input matrix: randoms, NaN-laced, outliers, ties."""
    rng = np.random.default_rng(7)
    cases = []
    for n, d in ((5, 7), (8, 100), (8, 1000), (11, 257), (16, 64),
                 (24, 301), (32, 129), (64, 67)):
        cases.append((f"normal_n{n}_d{d}",
                      rng.standard_normal((n, d)).astype(np.float32), 2))
    g = rng.standard_normal((8, 64)).astype(np.float32)
    g[1, 7] = np.nan
    cases.append(("nan_row", g, 2))
    g = rng.standard_normal((11, 40)).astype(np.float32)
    g[1] = np.nan
    cases.append(("nan_full_row_n11", g, 2))
    g = rng.standard_normal((11, 40)).astype(np.float32)
    g[1] = np.nan
    g[4] = np.nan
    cases.append(("nan_two_rows_n11", g, 2))
    g = rng.standard_normal((8, 64)).astype(np.float32) * 0.01 + 1
    g[3] = 1e6
    cases.append(("outlier", g, 1))
    g = np.ones((8, 16), dtype=np.float32)
    cases.append(("all_ties", g, 2))
    g = rng.standard_normal((8, 32)).astype(np.float32) * 1e6
    cases.append(("large_scale", g, 2))
    return cases


def main():
    from aggregathor_amd.ops import reference as R
    if not REF.exists():
        print("reference mount missing; skipping")
        return 0
    ref_ops, libpath = build_oracles()
    lib = load_lib(libpath)

    failures = []
    results = []

    def check(name, mine, ref, exact=False, rtol=3e-6, atol=1e-6):
        both_nan = np.isnan(mine) & np.isnan(ref)
        a = np.where(both_nan, 0, mine)
        b = np.where(both_nan, 0, ref)
        if not np.array_equal(np.isnan(mine), np.isnan(ref)):
            failures.append(f"{name}: NaN pattern mismatch")
            return
        if exact:
            ok = np.array_equal(a, b)
            err = float(np.abs(a - b).max()) if a.size else 0.0
        else:
            err = float(np.abs(a - b).max() /
                        max(np.abs(b).max(), 1e-30)) if a.size else 0.0
            ok = np.allclose(a, b, rtol=rtol, atol=atol * max(1.0, float(np.abs(b).max())))
        results.append((name, ok, err))
        if not ok:
            failures.append(f"{name}: max rel err {err:.3e}")

    for name, g, f in make_cases():
        n = g.shape[0]
        m = n - f - 2
        t = torch.from_numpy(g.copy())
        # Multi-Krum vs the reference's compiled op_krum kernel.
        check(f"krum/{name}", R.krum(t, f, m).numpy(),
              ref_op(ref_ops, 0, g, f, m))
        # Bulyan vs op_bulyan (valid configs only; NaN-laced inputs match
        # the reference binary to 1 ulp as well -- the documented
        # eviction-decrement refinement only differs on pathological
        # inf-pruned-distance inputs).
        if n >= 4 * f + 3:
            check(f"bulyan/{name}", R.bulyan(t, f, m).numpy(),
                  ref_op(ref_ops, 1, g, f, m))
        # Coordinate-wise rules vs the reference's ctypes library.
        check(f"median/{name}", R.median(t).numpy(), lib_median(lib, g),
              exact=("ties" not in name and "nan" not in name))
        beta = n - f
        if "nan" not in name:  # averaged_median's ref uses raw nth_element on NaN
            check(f"avgmed/{name}", R.averaged_median(t, beta).numpy(),
                  lib_averaged_median(lib, g, beta))
        check(f"avgnan/{name}", R.average_nan(t).numpy(),
              lib_average_nan(lib, g))
        # squared_distance on the first pair.
        mine_sd = float(R.pairwise_sqdist(t)[0, 1])
        ref_sd = lib_sqdist(lib, g[0], g[1])
        if np.isnan(mine_sd) and np.isnan(ref_sd):
            rel, ok = 0.0, True  # NaN inputs -> NaN distance on both sides
        else:
            rel = abs(mine_sd - ref_sd) / max(abs(ref_sd), 1e-30)
            ok = rel < 3e-6
        results.append((f"sqdist/{name}", ok, rel))
        if not ok:
            failures.append(f"sqdist/{name}: rel {rel:.3e}")

    print(f"{len(results)} checks, {len(failures)} failures")
    for name, ok, err in results:
        print(f"  {'OK ' if ok else 'FAIL'} {name:28s} err={err:.3e}")
    if failures:
        print("FAILURES:")
        for f_ in failures:
            print(" ", f_)
        return 1
    return 0


if __name__ == "__main__":
    sys.exit(main())
