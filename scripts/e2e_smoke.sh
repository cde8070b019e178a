#!/bin/bash
# End-to-end smoke: toy corpus -> brief training -> multi-process beam
# decode -> UNK replacement -> ROUGE (the reference's test.sh chain).
set -e
cd "$(dirname "$0")/.."

python - <<'EOF'
from nats_amd.data.synthetic import make_toy_corpus
make_toy_corpus("data")
EOF
mkdir -p models

FINISH=${FINISH:-5} DIM=${DIM:-32} python - <<'EOF'
import os
from nats_amd.engine.trainer import train
finish = int(os.environ.get("FINISH", "5"))
dim = int(os.environ.get("DIM", "32"))
train(dim_word=dim // 2, dim=dim, dim_att=max(8, dim // 4), n_words=64,
      maxlen=50, batch_size=8, valid_batch_size=8,
      saveto="models/model.npz",
      datasets=["data/toy_train_input.txt", "data/toy_train_output.txt"],
      valid_datasets=["data/toy_validation_input.txt",
                      "data/toy_validation_output.txt"],
      dictionary="data/toy_train_input.txt.pkl",
      validFreq=max(finish // 3, 5), saveFreq=max(finish // 3, 5),
      sampleFreq=10 ** 9, dispFreq=max(finish // 10, 1),
      finish_after=finish, clip_c=1.0, seed=3)
EOF

python scripts/gen.py -n -p ${NPROC:-2} -k 5 models/model.npz \
  data/toy_train_input.txt.pkl data/toy_test_input.txt temp.txt
python scripts/replace_unk.py data/toy_test_input.txt temp.txt final.txt
python scripts/rouge.py 1 N data/toy_test_output.txt final.txt
python scripts/rouge.py 2 N data/toy_test_output.txt final.txt
python scripts/rouge.py 1 L data/toy_test_output.txt final.txt
echo "e2e smoke OK"
