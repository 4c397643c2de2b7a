"""YAML training-config -> checkpoint-converter JSON (reference
scripts/yaml_converter.py:1-39 parity).

Takes an NxDT-style training YAML and emits the minimal model-config JSON
the checkpoint converter consumes (head/layer/hidden geometry, plus the
expert count for MoE checkpoints)."""

import argparse
import json

import yaml


def load_yaml_file(file_path):
    with open(file_path, "r") as f:
        return yaml.safe_load(f)


def convert_yaml_to_json(yaml_path, filename="yaml_config.json"):
    """Extract the geometry keys the checkpoint converter needs."""
    y = load_yaml_file(yaml_path)
    model = y["model"]
    cfg = {
        "num_hidden_layers": model["num_layers"],
        "num_attention_heads": model["num_attention_heads"],
        "hidden_size": model["hidden_size"],
        "num_key_value_heads": model["num_kv_heads"],
    }
    if "moe" in model:
        cfg["num_local_experts"] = model["moe"]["num_experts"]
    with open(filename, "w") as f:
        json.dump(cfg, f)
    return filename


def main():
    ap = argparse.ArgumentParser(
        description="Convert a training YAML to checkpoint-converter JSON")
    ap.add_argument("--yaml", required=True)
    ap.add_argument("--output", default="yaml_config.json")
    args = ap.parse_args()
    print(convert_yaml_to_json(args.yaml, args.output))


if __name__ == "__main__":
    main()
