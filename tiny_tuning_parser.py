#!/usr/bin/env python3
"""Scrape the final training loss from a JSON-line log (reference:
src/tiny_tuning_parser.py regex-scrapes worker prints; our logs are JSON)."""

import argparse
import json
import sys


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("logfile")
    p.add_argument("--lr", type=float, default=None)
    a = p.parse_args(argv)
    last = None
    with open(a.logfile) as f:
        for line in f:
            try:
                rec = json.loads(line)
            except json.JSONDecodeError:
                continue
            if rec.get("log") == "train" and "loss" in rec:
                last = rec
    if last is None:
        print(json.dumps({"lr": a.lr, "loss": None, "error": "no train records"}))
        return 1
    print(json.dumps({"lr": a.lr, "step": last["step"], "loss": last["loss"]}))
    return 0


if __name__ == "__main__":
    sys.exit(main())
