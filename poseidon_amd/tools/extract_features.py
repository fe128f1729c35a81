"""`extract_features` CLI (reference tools/extract_features.cpp).

    python -m poseidon_amd.tools.extract_features \
        --model snap_iter_1000.caffemodel --net deploy.prototxt \
        --blobs fc7,fc8 --batches 10 --out /tmp/feats
"""

from poseidon_amd.utils.feature_extractor import extract_features_cli

if __name__ == "__main__":
    extract_features_cli()
