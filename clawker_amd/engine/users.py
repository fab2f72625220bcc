"""Sandbox user materialization.

The flagship harness config declares `user: agent`, but hostfs-overlay
sandboxes see the HOST's /etc/passwd, which has no such user. The
reference solves this at image-build time (Dockerfile.base.tmpl user
setup with the host UID/GID — SURVEY.md §2.3); here the engine writes
the user straight into the sandbox's overlay upper at create time:
zero image rebuilds, per-sandbox identity, and the workspace owner's
uid/gid so bind-mounted files stay writable (the reference's host-UID
trick, without idmap mounts).

Only the ns backend can do this (it owns the rootfs); the proc backend
has no private /etc and must degrade explicitly (orchestrator policy).
"""
from __future__ import annotations

import os
from pathlib import Path

# groups whose device nodes the agent needs on a ROCm node: /dev/kfd and
# /dev/dri/renderD* are root:render / root:video 0660 on standard hosts
GPU_GROUPS = ("render", "video")


def _read_db(upper: Path, lowerdirs: list[str], rel: str) -> str:
    """The file the sandbox will see at /etc/<rel>: upper wins, then the
    top-most lower layer that carries it ('/' = hostfs base)."""
    cand = upper / "etc" / rel
    if cand.is_file():
        return cand.read_text()
    for ld in lowerdirs:
        cand = Path(ld) / "etc" / rel
        try:
            if cand.is_file():
                return cand.read_text()
        except OSError:
            continue
    return ""


def materialize_user(name: str, upper: Path, lowerdirs: list[str],
                     uid_hint: int = 0, gid_hint: int = 0) -> tuple[int, int]:
    """Ensure `name` resolves inside the sandbox; returns (uid, gid).

    If the user already exists in the sandbox's passwd (a real image that
    created it at build time), nothing is written. Otherwise passwd/group
    copies land in the overlay upper with the new entries appended, the
    user is added to the GPU device groups, and /home/<name> is created
    owned by the new uid.
    """
    passwd = _read_db(upper, lowerdirs, "passwd")
    for line in passwd.splitlines():
        f = line.split(":")
        if len(f) >= 4 and f[0] == name:
            return int(f[2]), int(f[3])

    used_uids = set()
    for line in passwd.splitlines():
        f = line.split(":")
        if len(f) >= 3 and f[2].isdigit():
            used_uids.add(int(f[2]))
    group = _read_db(upper, lowerdirs, "group")
    used_gids = set()
    for line in group.splitlines():
        f = line.split(":")
        if len(f) >= 3 and f[2].isdigit():
            used_gids.add(int(f[2]))

    # workspace-owner hint keeps bind-mounted files writable; hint 0
    # (root-owned workspace / no hint) falls through to the first free id
    uid = uid_hint
    if uid <= 0:
        uid = 1000
        while uid in used_uids:
            uid += 1
    gid = gid_hint
    if gid <= 0:
        gid = uid if uid not in used_gids else 1000
        while gid in used_gids:
            gid += 1

    etc = upper / "etc"
    etc.mkdir(parents=True, exist_ok=True)
    if not passwd.endswith("\n") and passwd:
        passwd += "\n"
    passwd += f"{name}:x:{uid}:{gid}::/home/{name}:/bin/sh\n"
    (etc / "passwd").write_text(passwd)
    os.chmod(etc / "passwd", 0o644)

    glines = group.splitlines()
    out = []
    have_own = False
    for line in glines:
        f = line.split(":")
        if len(f) >= 3 and f[0] in GPU_GROUPS:
            members = [m for m in f[3].split(",") if m] if len(f) > 3 else []
            if name not in members:
                members.append(name)
            line = ":".join(f[:3] + [",".join(members)])
        if len(f) >= 3 and f[0] == name:
            have_own = True
        out.append(line)
    if not have_own:
        out.append(f"{name}:x:{gid}:")
    (etc / "group").write_text("\n".join(out) + "\n")
    os.chmod(etc / "group", 0o644)

    home = upper / "home" / name
    home.mkdir(parents=True, exist_ok=True)
    os.chown(home, uid, gid)
    os.chmod(home, 0o755)
    try:
        os.chown(upper / "home", 0, 0)
    except OSError:
        pass
    return uid, gid


def is_named_user(user: str) -> bool:
    """True for users that need passwd resolution ('agent'), False for
    root/empty/numeric 'uid:gid' specs ckd resolves without a database."""
    if not user or user == "root":
        return False
    return not (set(user) <= set("0123456789:"))
