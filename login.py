"""wandb login helper (reference login.py role): logs this host into wandb.

    python login.py --key <API_KEY>

On multi-node setups run once per host (the reference looped over TPU
hosts; MI355X nodes are single-host-8-GPU, so one login per node).
"""
import argparse


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--key", required=True)
    args = p.parse_args()
    import wandb

    wandb.login(key=args.key)
    print("wandb login OK")


if __name__ == "__main__":
    main()
