#!/bin/bash
# COCO 2017 download helper (reference data/dataset/get_dataset.sh equivalent).
# This build/CI environment has NO network egress — run this on a machine with
# internet access, then point --ann/--img (scripts/build_dataset.py) at it.
set -euo pipefail
DEST=${1:-./coco2017}
mkdir -p "$DEST"
cd "$DEST"
echo "Downloading COCO 2017 train/val images + keypoint annotations into $PWD"
for f in train2017.zip val2017.zip annotations_trainval2017.zip; do
  if [ ! -f "$f" ]; then
    curl -LO "http://images.cocodataset.org/zips/$f" ||
    curl -LO "http://images.cocodataset.org/annotations/$f"
  fi
done
unzip -n train2017.zip
unzip -n val2017.zip
unzip -n annotations_trainval2017.zip
echo "Build the training h5 with:"
echo "  python -c 'from improved_body_parts_amd.data.coco import build_coco_h5;" \
     "build_coco_h5(\"$DEST/annotations/person_keypoints_train2017.json\"," \
     "\"$DEST/train2017\", \"coco_train_dataset512.h5\")'"
