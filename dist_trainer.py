#!/usr/bin/env python3
"""Alias entry point matching the reference's `dist_trainer.py` name
(SURVEY.md L7).  All flags identical to train.py."""
from train import main

if __name__ == "__main__":
    main()
