#!/usr/bin/env python3
"""Render etc/config.json from a template + metadata JSON.

The reference's config is rendered by the SAPI config-agent from a
mustache template (sapi_manifests/binder/template). This is the
standalone equivalent: {{KEY}} substitution from a metadata file.

usage: render-config.py template.json.in metadata.json > config.json
"""
import json
import re
import sys

template = open(sys.argv[1]).read()
meta = json.load(open(sys.argv[2]))

def sub(m):
    key = m.group(1).strip()
    v = meta.get(key, "")
    return json.dumps(v)[1:-1] if isinstance(v, str) else json.dumps(v)

out = re.sub(r"\{\{([^}]+)\}\}", sub, template)
json.loads(out)  # validate
sys.stdout.write(out)
