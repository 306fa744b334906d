"""Context options — the reference's flat `tuplex.*` KV store.

Reference: core/src/ContextOptions.cc:198-250 (release defaults). The same keys are
accepted (unknown tuplex.* keys are stored but unused); `tuplex.gpu.*` extensions
configure the MI355X backend.
"""
from typing import Optional

_SIZE_SUFFIX = {"KB": 1 << 10, "MB": 1 << 20, "GB": 1 << 30, "B": 1}


def parse_size(v) -> int:
    if isinstance(v, int):
        return v
    s = str(v).strip().upper()
    for suf in ("KB", "MB", "GB", "B"):
        if s.endswith(suf):
            return int(float(s[:-len(suf)])) * _SIZE_SUFFIX[suf]
    return int(s)


DEFAULTS = {
    "tuplex.useLLVMOptimizer": "true",     # accepted for compat; no LLVM here
    "tuplex.backend": "gpu",               # MI355X-native local backend
    "tuplex.partitionSize": "32MB",        # ContextOptions.cc:202
    "tuplex.runTimeMemory": "128MB",
    "tuplex.driverMemory": "1GB",
    "tuplex.executorMemory": "1GB",
    "tuplex.executorCount": "0",
    "tuplex.autoUpcast": "false",
    "tuplex.allowUndefinedBehavior": "false",
    "tuplex.csv.quotechar": '"',
    "tuplex.csv.separators": "[',', ';', '|', '\t']",
    "tuplex.csv.comments": "['#', '~']",
    "tuplex.csv.maxDetectionRows": "10000",
    "tuplex.csv.maxDetectionMemory": "256KB",
    "tuplex.normalcaseThreshold": "0.9",   # ContextOptions.cc:216
    "tuplex.optionalThreshold": "0.7",
    "tuplex.csv.selectionPushdown": "true",
    "tuplex.readBufferSize": "128KB",
    "tuplex.inputSplitSize": "64MB",       # ContextOptions.cc:227
    "tuplex.optimizer.generateParser": "false",
    "tuplex.optimizer.nullValueOptimization": "false",
    "tuplex.optimizer.filterPushdown": "true",
    "tuplex.optimizer.mergeExceptionsInOrder": "true",
    "tuplex.resolveWithInterpreterOnly": "false",
    "tuplex.webui.enable": "false",
    "tuplex.redirectToPythonLogging": "false",
    # MI355X extensions
    "tuplex.gpu.deviceCount": "1",
    "tuplex.gpu.device": "0",
    "tuplex.gpu.stagingBufferSize": "256MB",
    # GPU range granularity: the reference splits at 64MB for CPU-thread
    # work distribution (ContextOptions.cc:227); one MI355X wants fewer,
    # larger ranges (the C-ABI pipelines 4 sub-chunks on 2 streams per call)
    "tuplex.gpu.inputSplitSize": "256MB",
    "tuplex.gpu.heapFactor": "2.0",
    "tuplex.gpu.devices": "1",            # in-process device fan-out (N>1 opt-in; 0 = all visible)
    "tuplex.gpu.resolveProcesses": "0",   # parallel host resolver width (0 = auto, 1 = inline)
}


class Options:
    def __init__(self, conf: Optional[dict] = None):
        self.store = dict(DEFAULTS)
        self.explicit = set()
        if conf:
            for k, v in conf.items():
                key = k if k.startswith("tuplex.") else "tuplex." + k
                self.store[key] = self._fmt(v)
                self.explicit.add(key)

    def is_set(self, key) -> bool:
        key = key if key.startswith("tuplex.") else "tuplex." + key
        return key in getattr(self, "explicit", set())

    @staticmethod
    def _fmt(v):
        if isinstance(v, bool):
            return "true" if v else "false"
        return str(v)

    def get(self, key, default=None):
        key = key if key.startswith("tuplex.") else "tuplex." + key
        return self.store.get(key, default)

    def get_bool(self, key) -> bool:
        return str(self.get(key)).lower() == "true"

    def get_float(self, key) -> float:
        return float(self.get(key))

    def get_int(self, key) -> int:
        return int(self.get(key))

    @property
    def optional_threshold(self) -> float:
        return self.get_float("tuplex.optionalThreshold")

    @property
    def normalcase_threshold(self) -> float:
        return self.get_float("tuplex.normalcaseThreshold")

    @property
    def partition_size(self) -> int:
        return parse_size(self.get("tuplex.partitionSize"))

    @property
    def input_split_size(self) -> int:
        return parse_size(self.get("tuplex.inputSplitSize"))

    @property
    def gpu_input_split_size(self) -> int:
        return parse_size(self.get("tuplex.gpu.inputSplitSize"))

    @property
    def merge_in_order(self) -> bool:
        return self.get_bool("tuplex.optimizer.mergeExceptionsInOrder")

    def as_dict(self):
        return dict(self.store)
