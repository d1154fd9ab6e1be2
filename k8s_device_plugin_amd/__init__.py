"""MI355X-native Kubernetes vGPU sharing stack.

A from-scratch AMD MI355X (gfx950 / CDNA4) re-design of the capabilities of
4paradigm/k8s-device-plugin (OpenAIOS vGPU / Project-HAMi predecessor):

- ``plugin``     : per-node device plugin speaking the kubelet Device Plugin
                   gRPC API; enumerates MI355X GPUs from KFD sysfs and fans each
                   physical GPU out into fractional ``amd.com/gpu`` devices.
- ``scheduler``  : scheduler-extender (filter/bind) + mutating webhook doing
                   cluster-wide bin-packing of (GPU count, HBM MiB, CU %).
- ``monitor``    : node sidecar exporting Prometheus metrics from the
                   per-container shared-memory regions and writing back
                   priority/blocking feedback.
- ``ops``        : ctypes bindings for the native enforcement library
                   ``libvgpu-hip.so`` (LD_PRELOAD HIP interceptor, C).
- ``parallel``   : xGMI / NUMA topology discovery from KFD for
                   topology-aware placement.
- ``models``     : PyTorch-ROCm ai-benchmark model zoo used by ``bench.py``.
- ``utils``      : the wire protocol shared by all components — annotation
                   string codec, node lock, pending-pod matcher, k8s client.

The control plane is deliberately "annotations are the database": every
scheduling decision and node inventory lives in node/pod annotations, so any
component can crash and rebuild its state from the API server (reference
behavior: /root/reference/pkg/scheduler/scheduler.go:112-126).
"""

__version__ = "0.1.0"

# MI355X (gfx950, CDNA4) hardware constants used across the stack.
MI355X_NUM_CUS = 256          # 8 XCDs x 32 CUs
MI355X_NUM_XCDS = 8
MI355X_CUS_PER_XCD = 32
MI355X_HBM_MIB = 294912       # 288 GiB HBM3E
MI355X_DEVICE_TYPE = "AMD-Instinct-MI355X"
