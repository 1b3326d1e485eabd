"""Reference cfg-file compatibility: the flat KEY:VALUE format parsed by
InputInfo::readFromCfgFile (/root/reference/core/GraphSegment.cpp:222-292),
kept verbatim so the reference's own cfg files (gcn_cora.cfg, gcn_reddit.cfg,
...) drive this framework unchanged.

Recognized keys (superset ignored): ALGORITHM, VERTICES, LAYERS
("602-128-41"), EPOCHS, EDGE_FILE, FEATURE_FILE, LABEL_FILE, MASK_FILE,
PROC_OVERLAP, PROC_LOCAL, PROC_CUDA, PROC_REP, LOCK_FREE, OPTIM_KERNEL,
LEARN_RATE, WEIGHT_DECAY, DECAY_RATE, DECAY_EPOCH, DROP_RATE, BATCH_SIZE,
FANOUT.
"""
from dataclasses import dataclass, field
from typing import List


@dataclass
class InputInfo:
    algorithm: str = ""
    vertices: int = 0
    layer_string: str = ""
    epochs: int = 0
    edge_file: str = ""
    feature_file: str = ""
    label_file: str = ""
    mask_file: str = ""
    process_overlap: bool = False
    process_local: bool = False
    with_cuda: bool = False
    repthreshold: int = 0
    lock_free: bool = False
    optim_kernel_enable: bool = True
    learn_rate: float = 0.01
    weight_decay: float = 1e-4
    decay_rate: float = 0.97
    decay_epoch: int = 100
    drop_rate: float = 0.0
    batch_size: int = 0
    fanout: List[int] = field(default_factory=list)
    raw: dict = field(default_factory=dict)

    @property
    def layer_sizes(self) -> List[int]:
        """LAYERS '602-128-41' -> [602, 128, 41]."""
        return [int(x) for x in self.layer_string.split("-") if x]


def read_cfg(path: str) -> InputInfo:
    info = InputInfo()
    with open(path) as fh:
        for line in fh:
            line = line.strip()
            if not line or line.startswith("#") or ":" not in line:
                continue
            key, _, val = line.partition(":")
            key, val = key.strip(), val.strip()
            info.raw[key] = val
            if key == "ALGORITHM":
                info.algorithm = val
            elif key == "VERTICES":
                info.vertices = int(val)
            elif key == "LAYERS":
                info.layer_string = val
            elif key == "EPOCHS":
                info.epochs = int(val)
            elif key == "EDGE_FILE":
                info.edge_file = val
            elif key == "FEATURE_FILE":
                info.feature_file = val
            elif key == "LABEL_FILE":
                info.label_file = val
            elif key == "MASK_FILE":
                info.mask_file = val
            elif key == "PROC_OVERLAP":
                info.process_overlap = val not in ("0", "")
            elif key == "PROC_LOCAL":
                info.process_local = val not in ("0", "")
            elif key == "PROC_CUDA":
                info.with_cuda = val not in ("0", "")
            elif key == "PROC_REP":
                info.repthreshold = int(val)
            elif key == "LOCK_FREE":
                info.lock_free = val not in ("0", "")
            elif key == "OPTIM_KERNEL":
                # quirk kept visible: the reference parser sets this true
                # regardless of the value (GraphSegment.cpp:281-285); we
                # parse it honestly but nothing dispatches on it (one kernel
                # handles every feature width)
                info.optim_kernel_enable = val not in ("0", "")
            elif key == "LEARN_RATE":
                info.learn_rate = float(val)
            elif key == "WEIGHT_DECAY":
                info.weight_decay = float(val)
            elif key == "DECAY_RATE":
                info.decay_rate = float(val)
            elif key == "DECAY_EPOCH":
                info.decay_epoch = int(val)
            elif key == "DROP_RATE":
                info.drop_rate = float(val)
            elif key == "BATCH_SIZE":
                info.batch_size = int(val)
            elif key == "FANOUT":
                info.fanout = [int(x) for x in val.replace(",", " ").split()]
    return info
