export DEBIAN_FRONTEND=noninteractive
id -u agent >/dev/null 2>&1 || (groupadd -g GID agent 2>/dev/null || true; useradd -m -u UID -g GID -s /bin/bash agent || useradd -m -s /bin/bash agent)
(command -v python3) || echo 'clawker: best-effort step failed: ''command -v python3'
(apt-get update && apt-get install -y jq python3 python3-pip) || echo 'clawker: package install skipped (no network?)'
echo custom-step