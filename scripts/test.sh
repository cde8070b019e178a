#!/bin/bash -x
# Decode + UNK-replace + ROUGE pipeline — parity with the reference's
# test.sh:1-27 (same KL/CTX/STATE knobs, same 3-stage chain).
set -e
cd "$(dirname "$0")/.."

KL=${KL:-0}
CTX=${CTX:-0}
STATE=${STATE:-0}
ROOT=${ROOT:-.}
MODEL=${MODEL:-$ROOT/models/model.npz}
DIC=${DIC:-$ROOT/data/toy_train_input.txt.pkl}
INPUT=${INPUT:-$ROOT/data/toy_test_input.txt}
TEMP=${TEMP:-./temp.txt}
GEN=${GEN:-./final.txt}
REF=${REF:-$ROOT/data/toy_test_output.txt}

# bootstrap the toy corpus on a fresh clone (deterministic, seed=1234)
python -c "import sys; sys.path.insert(0, 'scripts'); \
  from train_nats import ensure_toy_corpus; ensure_toy_corpus()"

# generate summaries (beam k=5, 10 workers, length-normalized)
python scripts/gen.py -n -p 10 -k 5 -l ${KL} -x ${CTX} -s ${STATE} $MODEL $DIC $INPUT $TEMP

# replace unk
python scripts/replace_unk.py $INPUT $TEMP $GEN

# calculate rouge score
python scripts/rouge.py 1 N $REF $GEN
python scripts/rouge.py 2 N $REF $GEN
python scripts/rouge.py 1 L $REF $GEN
