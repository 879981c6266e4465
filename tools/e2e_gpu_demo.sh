#!/bin/bash
# Full CLI path on a GPU box: synthesize Shifu-style CSV, train via
# shifu_amd.run (launcher -> rank -> native ingest -> HIP kernels -> export),
# then score through the serving layer.  Writes artifacts under gpurun_out/.
set -e
cd /root/repo
OUT=${E2E_OUT:-/tmp/e2e_demo}
mkdir -p $OUT
python - <<'PY'
import json, sys
sys.path.insert(0, "/root/repo")
from shifu_amd.data.synthetic import generate_synthetic_csv
paths = generate_synthetic_csv("/tmp/e2e_demo/data", n_rows=500_000,
                               n_dense=50, vocab_sizes=[100_000]*8, n_files=16, seed=42)
print(f"generated {len(paths)} files")
cc = [{"columnNum": 0, "columnFlag": "Target"}, {"columnNum": 1, "columnFlag": "Weight"}]
cc += [{"columnNum": i, "finalSelect": True, "columnType": "N"} for i in range(2, 52)]
cc += [{"columnNum": i, "finalSelect": True, "columnType": "C", "vocabSize": 100_000}
       for i in range(52, 60)]
json.dump(cc, open("/tmp/e2e_demo/ColumnConfig.json", "w"))
json.dump({"train": {"numTrainEpochs": 2, "validSetRate": 0.1,
                     "params": {"NumHiddenLayers": 3, "NumHiddenNodes": [512, 256, 128],
                                "ActivationFunc": ["relu", "relu", "relu"],
                                "LearningRate": 0.001, "Optimizer": "adam",
                                "Loss": "sigmoid_ce", "MiniBatchSize": 8192, "L2Reg": 0.0}}},
          open("/tmp/e2e_demo/ModelConfig.json", "w"))
json.dump({"num_gpus": 1, "training_data_path": ["/tmp/e2e_demo/data"],
           "tmp_model_path": "/tmp/e2e_demo/ckpt",
           "final_model_path": "/tmp/e2e_demo/final",
           "log_dir": "/tmp/e2e_demo/logs",
           "model_type": "wide_deep", "embed_dim": 16, "enable_trace": False},
          open("/tmp/e2e_demo/run.json", "w"))
PY
time python -m shifu_amd.run --run-config $OUT/run.json \
    --model-config $OUT/ModelConfig.json --column-config $OUT/ColumnConfig.json
echo "--- board ---"; cat $OUT/logs/progress.board
echo "--- rank log tail ---"; tail -5 $OUT/logs/rank-0.log
echo "--- export ---"; ls $OUT/final
python - <<'PY'
import sys
sys.path.insert(0, "/root/repo")
from shifu_amd.serve import ShifuScorer
sc = ShifuScorer()
sc.init("/tmp/e2e_demo/final/GenericModelConfig.json")
row = [0.1] * 50 + [5, 17, 3, 99, 1000, 7, 42, 12345]
print("score:", sc.compute(row))
PY
