"""Incentive (credit) subsystem.
Reference parity: infomesh/credits/ (SURVEY.md §2.6)."""
