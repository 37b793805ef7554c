#!/bin/bash
export PYTHONPATH=$GRAFT_REPO_ROOT
for i in 1 2 3; do
  timeout 200 python - <<'PYEOF'
import torch
from torchsnapshot_amd import Snapshot
from torchsnapshot_amd.state_dict import StateDict
sd = StateDict(w=torch.randn(512, 512, device="cuda"))
snap = Snapshot.take("/tmp/t/snap", {"sd": sd})
snap.delete()
print("exit-clean-check done")
PYEOF
  echo "RC=$?"
done
