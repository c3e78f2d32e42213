#!/bin/bash
# ThreadSanitizer pass over the C++ core's CPU path.
#
# Builds a second copy of the extension with -fsanitize=thread (into
# tools/.tsan/, not the production _core.so), then runs a multi-threaded
# enqueue/synchronize stress under TSan. libtorch/python are not
# instrumented, so reports are filtered to frames in axon's own code
# (tools/tsan.supp suppresses the uninstrumented libraries).
set -u
cd "$(dirname "$0")/.."
REPO="$PWD"
OUT="$REPO/tools/.tsan"
mkdir -p "$OUT"

PY_INC=$(python3 -c "import sysconfig; print(sysconfig.get_paths()['include'])")
T_INC=$(python3 -c "import torch, os; print(os.path.dirname(torch.__file__) + '/include')")
T_LIB=$(python3 -c "import torch, os; print(os.path.dirname(torch.__file__) + '/lib')")
TSAN_RT=$(/opt/rocm/lib/llvm/bin/clang++ -print-file-name=libclang_rt.tsan-x86_64.so)

FLAGS="-fPIC -O1 -g -std=c++17 -fsanitize=thread -D__HIP_PLATFORM_AMD__=1 -DUSE_ROCM=1 \
  -D_GLIBCXX_USE_CXX11_ABI=1 -DTORCH_API_INCLUDE_EXTENSION_H -DTORCH_EXTENSION_NAME=_core \
  -D__HIP_NO_HALF_OPERATORS__=1 -D__HIP_NO_HALF_CONVERSIONS__=1 \
  --offload-arch=gfx950 \
  -I$T_INC -I$T_INC/torch/csrc/api/include -I$PY_INC -I/opt/rocm/include -I$REPO/axon/csrc"

objs=""
for src in axon/csrc/bindings.cpp axon/csrc/optim_ops.cpp axon/csrc/bn_ops.cpp \
           axon/csrc/core/*.cpp axon/csrc/kernels/*.hip; do
  obj="$OUT/$(echo "$src" | tr '/' '_').o"
  if [ ! -f "$obj" ] || [ "$src" -nt "$obj" ]; then
    echo "[tsan] CC $src"
    /opt/rocm/bin/hipcc -c "$src" -o "$obj" $FLAGS || exit 1
  fi
  objs="$objs $obj"
done

/opt/rocm/bin/hipcc -shared -fsanitize=thread -o "$OUT/_core.so" $objs \
  -L"$T_LIB" -ltorch -ltorch_cpu -ltorch_hip -ltorch_python -lc10 -lc10_hip -lrccl \
  -Wl,-rpath,"$T_LIB" -Wl,-rpath,/opt/rocm/lib || exit 1

# Stage a package overlay whose axon/_core.so is the TSan build.
PKG="$OUT/pkg"
rm -rf "$PKG"; mkdir -p "$PKG"
cp -r "$REPO/axon" "$PKG/axon"
cp "$OUT/_core.so" "$PKG/axon/_core.so"

export AXON_TSAN_REPO="$PKG"
export TSAN_OPTIONS="suppressions=$REPO/tools/tsan.supp report_bugs=1 \
  second_deadlock_stack=1 log_path=$OUT/report exitcode=66 history_size=7"
LD_PRELOAD="$TSAN_RT" python3 "$REPO/tools/tsan_stress.py"
rc=$?

# Count reports whose stacks touch axon code (uninstrumented-library noise
# is suppressed; anything left that names axon/csrc is a real finding).
hits=$(grep -l "axon" "$OUT"/report.* 2>/dev/null | wc -l)
total=$(ls "$OUT"/report.* 2>/dev/null | wc -l)
echo "[tsan] stress exit=$rc, reports=$total, axon-frame reports=$hits"
if [ "$rc" != "0" ] && [ "$rc" != "66" ]; then exit "$rc"; fi
[ "$hits" = "0" ] && echo "[tsan] PASS (no races in axon frames)" || exit 1
