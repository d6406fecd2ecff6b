#!/bin/sh
# Model loader: `load <src-url> <dest-dir>` — downloads model artifacts into
# the shared cache dir / adapter dir (reference analog:
# components/model-loader/load.sh; used by cache Jobs and the adapter
# sidecar in cluster deployments).
set -eu

SRC="${1:?usage: load <src-url> <dest>}"
DEST="${2:?usage: load <src-url> <dest>}"

mkdir -p "$DEST"
case "$SRC" in
  hf://*)
    REPO="${SRC#hf://}"
    huggingface-cli download "$REPO" --local-dir "$DEST" ;;
  s3://*)
    aws s3 sync "$SRC" "$DEST" ;;
  gs://*)
    gcloud storage rsync --recursive "$SRC" "$DEST" ;;
  oss://*)
    ossutil cp -r "$SRC" "$DEST" ;;
  file://*)
    cp -r "${SRC#file://}/." "$DEST" ;;
  *)
    echo "unsupported url scheme: $SRC" >&2
    exit 1 ;;
esac
echo "loaded $SRC -> $DEST"
