import sys, pathlib
sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
import torch
from waternet_amd.ops import ext
o = ext().probe_tr().cpu().numpy()
import numpy as np
np.set_printoptions(linewidth=200, suppress=True)
for lane in [0,1,2,3,4,15,16,17,31,32,48]:
    print(f"lane {lane:2d} f0:", o[lane,0], " f1:", o[lane,1])

raw = ext().probe_tr_raw().cpu().numpy()
print("RAW permutation (lane -> delivered element indices):")
for lane in range(64):
    print(f"lane {lane:2d}:", raw[lane])
