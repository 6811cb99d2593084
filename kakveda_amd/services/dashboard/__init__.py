"""Dashboard service: UI/API/auth front-end for the whole platform.

Route parity with the reference dashboard (SURVEY.md section 2.3,
reference services/dashboard/app.py): auth + RBAC, scenario runner,
runs/traces/spans with waterfall, warnings analytics, playground,
datasets/evals, prompts/experiments, agent registry, projects/API
keys/budgets, admin purge — backed by SQLite (SQLAlchemy) and the other
services over the shared Transport.
"""

from kakveda_amd.services.dashboard.app import create_app  # noqa: F401
