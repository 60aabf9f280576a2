#!/bin/bash
# SURVEY §5.2 sanitizer pass: run the GPU kernel test suite against the
# device-AddressSanitizer build of the ops extension (gfx950:xnack+,
# asanrtl.bc device runtime). Run once per round on a GPU box:
#
#   gpurun --timeout 1800 -- 'bash tools/gpu_sanitize.sh'
#
# The log lands in gpurun_out/sanitize_kernels.log; a clean pass (or the
# fixed findings) is copied into profiles/ for the round record. If the
# ASAN build cannot initialize on the box (ROCm ships no instrumented
# HSA/HIP runtime in this image), the script falls back to a serialized
# debug pass (AMD_SERIALIZE_KERNEL/COPY) over the same kernel tests so
# the round still records a memory-order/fault-surface check.
set -u
cd "$(dirname "$0")/.."
LOG=gpurun_out/sanitize_kernels.log
mkdir -p gpurun_out
: > "$LOG"

SO=mi355x_scale/ops/_C_asan.cpython-310-x86_64-linux-gnu.so
if [ ! -f "$SO" ]; then
  echo "building ASAN extension (separate temp dir)..." | tee -a "$LOG"
  MI355X_ASAN=1 python setup.py build_ext --inplace \
      --build-temp build/asan_temp >> "$LOG" 2>&1 || exit 1
fi

# preload the SAME asan runtime the extension is linked against (torch's
# BuildExtension links through g++, so this is gcc's libasan, which also
# serves the clang-instrumented host halves of the .hip files)
ASAN_RT=$(ldd "$SO" | awk '/libasan/ {print $3; exit}')
echo "asan runtime: $ASAN_RT" | tee -a "$LOG"

KTESTS="tests/test_batched_fit.py tests/test_fused_bn.py \
tests/test_maxpool.py tests/test_flat_adam.py tests/test_preprocess.py \
tests/test_stemconv.py tests/test_conv3x3_wrw.py"

export HSA_XNACK=1
export MI355X_OPS_EXT=_C_asan
export ASAN_OPTIONS=detect_leaks=0:halt_on_error=0:abort_on_error=0:verify_asan_link_order=0
echo "== probe: ASAN extension load + one kernel ==" | tee -a "$LOG"
LD_PRELOAD="$ASAN_RT" timeout 240 python - >> "$LOG" 2>&1 <<'EOF'
import torch
from mi355x_scale.ops import _C, HAVE_EXT
assert HAVE_EXT and "_C_asan" in _C.__file__, _C
x = torch.randint(0, 255, (4, 16, 16, 3), dtype=torch.uint8, device="cuda")
from mi355x_scale.ops import normalize_images
y = normalize_images(x)
torch.cuda.synchronize()
print("ASAN probe ok:", y.shape, y.dtype)
EOF
PROBE_RC=$?
echo "probe rc=$PROBE_RC" | tee -a "$LOG"
if [ $PROBE_RC -ne 0 ]; then
  # gcc-libasan's __cxa_throw interceptor needs the C++ runtime resolved
  # before torch's lazy dlopen chain — retry with libstdc++ pre-bound
  STDCXX=$(ldconfig -p | awk '/libstdc\+\+\.so\.6 \(/ {print $NF; exit}')
  echo "retry with libstdc++ preloaded: $STDCXX" | tee -a "$LOG"
  LD_PRELOAD="$STDCXX:$ASAN_RT" timeout 240 python - >> "$LOG" 2>&1 <<'EOF'
import torch
from mi355x_scale.ops import _C, HAVE_EXT
assert HAVE_EXT and "_C_asan" in _C.__file__, _C
x = torch.randint(0, 255, (4, 16, 16, 3), dtype=torch.uint8, device="cuda")
from mi355x_scale.ops import normalize_images
y = normalize_images(x)
torch.cuda.synchronize()
print("ASAN probe ok (stdc++ preload):", y.shape, y.dtype)
EOF
  PROBE_RC=$?
  echo "probe2 rc=$PROBE_RC" | tee -a "$LOG"
  [ $PROBE_RC -eq 0 ] && ASAN_RT="$STDCXX:$ASAN_RT"
fi

if [ $PROBE_RC -eq 0 ]; then
  echo "== device-ASAN kernel suite ($(date -u +%FT%TZ)) ==" | tee -a "$LOG"
  LD_PRELOAD="$ASAN_RT" timeout 1200 python -m pytest $KTESTS -m gpu -q \
      >> "$LOG" 2>&1
  RC=$?
  MODE=asan
else
  echo "== FALLBACK: serialized debug pass (no usable device ASAN) ==" \
      | tee -a "$LOG"
  unset MI355X_OPS_EXT
  AMD_SERIALIZE_KERNEL=3 AMD_SERIALIZE_COPY=3 HSA_XNACK=1 \
      timeout 1200 python -m pytest $KTESTS -m gpu -q >> "$LOG" 2>&1
  RC=$?
  MODE=serialize
fi
echo "mode=$MODE pytest rc=$RC" | tee -a "$LOG"
grep -E "ERROR: AddressSanitizer|SUMMARY: AddressSanitizer|heap-buffer|global-buffer|device-malloc" "$LOG" \
  && echo "ASAN FINDINGS ABOVE" | tee -a "$LOG" \
  || echo "no ASAN reports in log" | tee -a "$LOG"
tail -6 "$LOG"
exit $RC
