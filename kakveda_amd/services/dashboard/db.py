"""Dashboard persistence: SQLAlchemy models over SQLite.

Model parity with the reference ORM (reference services/dashboard/db.py:
25-361 — 22 tables: users/roles/auth, projects/keys/budgets, agent
registry, scenario runs, warning events, trace runs + nested spans,
prompt library/versions, experiments, datasets/examples, feedback,
evaluation runs/results). Implemented fresh on SQLAlchemy 2.0 declarative
style; ``init_db`` is idempotent (create_all covers the reference's
hand-rolled ALTER-based migrate_db since this schema starts complete).
"""

from __future__ import annotations

import datetime as dt
import os
from typing import Optional

from sqlalchemy import (
    Boolean,
    DateTime,
    Float,
    ForeignKey,
    Integer,
    String,
    Text,
    create_engine,
)
from sqlalchemy.orm import DeclarativeBase, Mapped, mapped_column, sessionmaker


def utcnow() -> dt.datetime:
    return dt.datetime.now(dt.timezone.utc)


class Base(DeclarativeBase):
    pass


class User(Base):
    __tablename__ = "users"
    id: Mapped[int] = mapped_column(Integer, primary_key=True)
    email: Mapped[str] = mapped_column(String(255), unique=True, index=True)
    password_hash: Mapped[str] = mapped_column(String(512))
    display_name: Mapped[str] = mapped_column(String(255), default="")
    is_active: Mapped[bool] = mapped_column(Boolean, default=True)
    created_at: Mapped[dt.datetime] = mapped_column(DateTime, default=utcnow)


class Role(Base):
    __tablename__ = "roles"
    id: Mapped[int] = mapped_column(Integer, primary_key=True)
    name: Mapped[str] = mapped_column(String(64), unique=True)


class UserRole(Base):
    __tablename__ = "user_roles"
    id: Mapped[int] = mapped_column(Integer, primary_key=True)
    user_id: Mapped[int] = mapped_column(ForeignKey("users.id"), index=True)
    role_id: Mapped[int] = mapped_column(ForeignKey("roles.id"), index=True)


class PasswordResetToken(Base):
    __tablename__ = "password_reset_tokens"
    id: Mapped[int] = mapped_column(Integer, primary_key=True)
    user_id: Mapped[int] = mapped_column(ForeignKey("users.id"), index=True)
    token: Mapped[str] = mapped_column(String(128), unique=True)
    expires_at: Mapped[dt.datetime] = mapped_column(DateTime)
    used: Mapped[bool] = mapped_column(Boolean, default=False)


class AuditEvent(Base):
    __tablename__ = "audit_events"
    id: Mapped[int] = mapped_column(Integer, primary_key=True)
    ts: Mapped[dt.datetime] = mapped_column(DateTime, default=utcnow, index=True)
    actor: Mapped[str] = mapped_column(String(255), default="")
    action: Mapped[str] = mapped_column(String(128), index=True)
    target: Mapped[str] = mapped_column(String(255), default="")
    meta_json: Mapped[str] = mapped_column(Text, default="{}")


class Project(Base):
    __tablename__ = "projects"
    id: Mapped[int] = mapped_column(Integer, primary_key=True)
    name: Mapped[str] = mapped_column(String(255), unique=True)
    description: Mapped[str] = mapped_column(Text, default="")
    created_at: Mapped[dt.datetime] = mapped_column(DateTime, default=utcnow)


class ProjectMember(Base):
    __tablename__ = "project_members"
    id: Mapped[int] = mapped_column(Integer, primary_key=True)
    project_id: Mapped[int] = mapped_column(ForeignKey("projects.id"), index=True)
    user_id: Mapped[int] = mapped_column(ForeignKey("users.id"), index=True)
    role: Mapped[str] = mapped_column(String(64), default="member")


class ProjectApiKey(Base):
    __tablename__ = "project_api_keys"
    id: Mapped[int] = mapped_column(Integer, primary_key=True)
    project_id: Mapped[int] = mapped_column(ForeignKey("projects.id"), index=True)
    name: Mapped[str] = mapped_column(String(255), default="default")
    key_hash: Mapped[str] = mapped_column(String(128), index=True)  # sha256
    created_at: Mapped[dt.datetime] = mapped_column(DateTime, default=utcnow)
    revoked: Mapped[bool] = mapped_column(Boolean, default=False)


class ProjectBudget(Base):
    __tablename__ = "project_budgets"
    id: Mapped[int] = mapped_column(Integer, primary_key=True)
    project_id: Mapped[int] = mapped_column(ForeignKey("projects.id"), unique=True)
    monthly_usd_micro: Mapped[int] = mapped_column(Integer, default=0)  # 0 = unlimited


class AgentRegistry(Base):
    __tablename__ = "agent_registry"
    id: Mapped[int] = mapped_column(Integer, primary_key=True)
    name: Mapped[str] = mapped_column(String(255), unique=True)
    base_url: Mapped[str] = mapped_column(String(512))
    enabled: Mapped[bool] = mapped_column(Boolean, default=True)
    auth_env_var: Mapped[str] = mapped_column(String(255), default="")
    capabilities_json: Mapped[str] = mapped_column(Text, default="[]")
    last_heartbeat: Mapped[Optional[dt.datetime]] = mapped_column(DateTime, nullable=True)
    registered_by: Mapped[str] = mapped_column(String(255), default="")
    created_at: Mapped[dt.datetime] = mapped_column(DateTime, default=utcnow)


class ScenarioRun(Base):
    __tablename__ = "scenario_runs"
    id: Mapped[int] = mapped_column(Integer, primary_key=True)
    ts: Mapped[dt.datetime] = mapped_column(DateTime, default=utcnow, index=True)
    app_id: Mapped[str] = mapped_column(String(255), index=True)
    prompt: Mapped[str] = mapped_column(Text)
    response: Mapped[str] = mapped_column(Text, default="")
    warn_action: Mapped[str] = mapped_column(String(32), default="")
    warn_confidence: Mapped[float] = mapped_column(Float, default=0.0)
    trace_id: Mapped[str] = mapped_column(String(255), default="")


class WarningEvent(Base):
    __tablename__ = "warning_events"
    id: Mapped[int] = mapped_column(Integer, primary_key=True)
    ts: Mapped[dt.datetime] = mapped_column(DateTime, default=utcnow, index=True)
    app_id: Mapped[str] = mapped_column(String(255), index=True)
    action: Mapped[str] = mapped_column(String(32))
    confidence: Mapped[float] = mapped_column(Float, default=0.0)
    pattern_id: Mapped[str] = mapped_column(String(64), default="")
    failure_id: Mapped[str] = mapped_column(String(64), default="")
    message: Mapped[str] = mapped_column(Text, default="")
    prompt: Mapped[str] = mapped_column(Text, default="")
    est_cost_usd_micro: Mapped[int] = mapped_column(Integer, default=0)


class TraceRun(Base):
    __tablename__ = "trace_runs"
    id: Mapped[int] = mapped_column(Integer, primary_key=True)
    ts: Mapped[dt.datetime] = mapped_column(DateTime, default=utcnow, index=True)
    trace_id: Mapped[str] = mapped_column(String(255), unique=True, index=True)
    app_id: Mapped[str] = mapped_column(String(255), index=True)
    project_id: Mapped[Optional[int]] = mapped_column(
        ForeignKey("projects.id"), nullable=True, index=True
    )
    provider: Mapped[str] = mapped_column(String(64), default="")
    model: Mapped[str] = mapped_column(String(255), default="")
    prompt: Mapped[str] = mapped_column(Text, default="")
    response: Mapped[str] = mapped_column(Text, default="")
    latency_ms: Mapped[float] = mapped_column(Float, default=0.0)
    tokens_in: Mapped[int] = mapped_column(Integer, default=0)
    tokens_out: Mapped[int] = mapped_column(Integer, default=0)
    cost_usd_micro: Mapped[int] = mapped_column(Integer, default=0)
    error: Mapped[str] = mapped_column(Text, default="")
    tags: Mapped[str] = mapped_column(String(512), default="")
    label: Mapped[str] = mapped_column(String(255), default="")
    source: Mapped[str] = mapped_column(String(64), default="event")


class TraceSpan(Base):
    __tablename__ = "trace_spans"
    id: Mapped[int] = mapped_column(Integer, primary_key=True)
    trace_run_id: Mapped[int] = mapped_column(ForeignKey("trace_runs.id"), index=True)
    parent_id: Mapped[Optional[int]] = mapped_column(
        ForeignKey("trace_spans.id"), nullable=True
    )
    name: Mapped[str] = mapped_column(String(255))
    start: Mapped[dt.datetime] = mapped_column(DateTime, default=utcnow)
    end: Mapped[Optional[dt.datetime]] = mapped_column(DateTime, nullable=True)
    duration_ms: Mapped[float] = mapped_column(Float, default=0.0)
    meta_json: Mapped[str] = mapped_column(Text, default="{}")


class PromptLibrary(Base):
    __tablename__ = "prompt_library"
    id: Mapped[int] = mapped_column(Integer, primary_key=True)
    name: Mapped[str] = mapped_column(String(255), unique=True)
    description: Mapped[str] = mapped_column(Text, default="")
    default_provider: Mapped[str] = mapped_column(String(64), default="")
    default_model: Mapped[str] = mapped_column(String(255), default="")
    tags: Mapped[str] = mapped_column(String(512), default="")
    created_at: Mapped[dt.datetime] = mapped_column(DateTime, default=utcnow)


class PromptVersion(Base):
    __tablename__ = "prompt_versions"
    id: Mapped[int] = mapped_column(Integer, primary_key=True)
    prompt_id: Mapped[int] = mapped_column(ForeignKey("prompt_library.id"), index=True)
    version: Mapped[int] = mapped_column(Integer)
    content: Mapped[str] = mapped_column(Text)
    created_at: Mapped[dt.datetime] = mapped_column(DateTime, default=utcnow)


class Experiment(Base):
    __tablename__ = "experiments"
    id: Mapped[int] = mapped_column(Integer, primary_key=True)
    name: Mapped[str] = mapped_column(String(255), unique=True)
    description: Mapped[str] = mapped_column(Text, default="")
    created_at: Mapped[dt.datetime] = mapped_column(DateTime, default=utcnow)


class ExperimentRun(Base):
    __tablename__ = "experiment_runs"
    id: Mapped[int] = mapped_column(Integer, primary_key=True)
    experiment_id: Mapped[int] = mapped_column(ForeignKey("experiments.id"), index=True)
    trace_run_id: Mapped[int] = mapped_column(ForeignKey("trace_runs.id"), index=True)


class Dataset(Base):
    __tablename__ = "datasets"
    id: Mapped[int] = mapped_column(Integer, primary_key=True)
    name: Mapped[str] = mapped_column(String(255), unique=True)
    description: Mapped[str] = mapped_column(Text, default="")
    created_at: Mapped[dt.datetime] = mapped_column(DateTime, default=utcnow)


class DatasetExample(Base):
    __tablename__ = "dataset_examples"
    id: Mapped[int] = mapped_column(Integer, primary_key=True)
    dataset_id: Mapped[int] = mapped_column(ForeignKey("datasets.id"), index=True)
    input_text: Mapped[str] = mapped_column(Text)
    expected: Mapped[str] = mapped_column(Text, default="")
    meta_json: Mapped[str] = mapped_column(Text, default="{}")


class RunFeedback(Base):
    __tablename__ = "run_feedback"
    id: Mapped[int] = mapped_column(Integer, primary_key=True)
    trace_run_id: Mapped[int] = mapped_column(ForeignKey("trace_runs.id"), index=True)
    thumb: Mapped[int] = mapped_column(Integer, default=0)  # -1 | 0 | 1
    label: Mapped[str] = mapped_column(String(255), default="")
    comment: Mapped[str] = mapped_column(Text, default="")
    ts: Mapped[dt.datetime] = mapped_column(DateTime, default=utcnow)


class EvaluationRun(Base):
    __tablename__ = "evaluation_runs"
    id: Mapped[int] = mapped_column(Integer, primary_key=True)
    ts: Mapped[dt.datetime] = mapped_column(DateTime, default=utcnow)
    dataset_id: Mapped[int] = mapped_column(ForeignKey("datasets.id"), index=True)
    name: Mapped[str] = mapped_column(String(255), default="")
    status: Mapped[str] = mapped_column(String(32), default="done")
    summary_json: Mapped[str] = mapped_column(Text, default="{}")


class EvaluationResult(Base):
    __tablename__ = "evaluation_results"
    id: Mapped[int] = mapped_column(Integer, primary_key=True)
    evaluation_run_id: Mapped[int] = mapped_column(
        ForeignKey("evaluation_runs.id"), index=True
    )
    example_id: Mapped[int] = mapped_column(ForeignKey("dataset_examples.id"))
    passed: Mapped[bool] = mapped_column(Boolean, default=False)
    score: Mapped[float] = mapped_column(Float, default=0.0)
    latency_ms: Mapped[float] = mapped_column(Float, default=0.0)
    output: Mapped[str] = mapped_column(Text, default="")
    detail_json: Mapped[str] = mapped_column(Text, default="{}")


def make_engine(db_path: Optional[str] = None, url: Optional[str] = None):
    """Engine from an explicit URL, the DATABASE_URL env (reference
    docker-compose.prod.yml runs the dashboard on postgres this way), or
    a local SQLite file path."""
    url = url or os.environ.get("DATABASE_URL") or ""
    if not url:
        url = f"sqlite:///{db_path}"
    kwargs = {}
    if url.startswith("sqlite"):
        kwargs["connect_args"] = {"check_same_thread": False}
    else:
        kwargs["pool_pre_ping"] = True
    return create_engine(url, **kwargs)


def _default_sql(col) -> str:
    """DEFAULT clause for ALTER TABLE ADD COLUMN, from the model default."""
    d = getattr(col.default, "arg", None)
    if d is None or callable(d):
        return ""
    if isinstance(d, bool):
        return f" DEFAULT {1 if d else 0}"
    if isinstance(d, (int, float)):
        return f" DEFAULT {d}"
    return " DEFAULT '{}'".format(str(d).replace("'", "''"))


def migrate_db(eng) -> list:
    """Idempotent startup migrations (reference services/dashboard/db.py:
    368-644 runs hand-rolled ALTERs on every boot): add any model column
    missing from an existing table via ALTER TABLE ADD COLUMN (with the
    model's scalar default), then create any missing tables. Safe to run
    on every startup; returns the DDL statements applied."""
    from sqlalchemy import inspect, text

    insp = inspect(eng)
    applied = []
    existing = set(insp.get_table_names())
    with eng.begin() as conn:
        for table in Base.metadata.sorted_tables:
            if table.name not in existing:
                continue  # create_all below creates it whole
            have = {c["name"] for c in insp.get_columns(table.name)}
            for col in table.columns:
                if col.name in have:
                    continue
                ddl = (
                    f"ALTER TABLE {table.name} ADD COLUMN {col.name} "
                    f"{col.type.compile(eng.dialect)}{_default_sql(col)}"
                )
                conn.execute(text(ddl))
                applied.append(ddl)
    Base.metadata.create_all(eng)
    return applied


def init_db(db_path: Optional[str] = None, url: Optional[str] = None):
    """Create (or migrate-forward) the schema; returns a session factory."""
    if db_path and not (url or os.environ.get("DATABASE_URL")):
        os.makedirs(os.path.dirname(db_path) or ".", exist_ok=True)
    eng = make_engine(db_path, url=url)
    migrate_db(eng)
    return sessionmaker(bind=eng, expire_on_commit=False)
