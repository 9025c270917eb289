"""Base project scaffold: the files the reference delegates to upstream
kubebuilder plugins (golang/v3 + kustomize/v1), re-implemented here
(SURVEY.md §7 hard part #3 — pkg/cli/init.go:27-45 bundles those plugins
before the workload plugin runs).

Covers: .gitignore, .dockerignore, hack/boilerplate.go.txt, the
config/ kustomize tree (default / manager / prometheus / rbac / crd),
and the per-API rbac editor/viewer roles.
"""

from __future__ import annotations

import os

from ..context import Context
from ..machinery import File, IfExists

APACHE2_BOILERPLATE = """/*
Copyright 2021.

Licensed under the Apache License, Version 2.0 (the "License");
you may not use this file except in compliance with the License.
You may obtain a copy of the License at

    http://www.apache.org/licenses/LICENSE-2.0

Unless required by applicable law or agreed to in writing, software
distributed under the License is distributed on an "AS IS" BASIS,
WITHOUT WARRANTIES OR CONDITIONS OF ANY KIND, either express or implied.
See the License for the specific language governing permissions and
limitations under the License.
*/"""


def boilerplate_file(content: str) -> File:
    return File(
        os.path.join("hack", "boilerplate.go.txt"),
        content + "\n",
        IfExists.SKIP,
    )


def gitignore(ctx: Context) -> File:
    content = """
# Binaries for programs and plugins
*.exe
*.exe~
*.dll
*.so
*.dylib
bin
testbin/*

# Test binary, build with `go test -c`
*.test

# Output of the go coverage tool, specifically when used with LiteIDE
*.out

# Kubernetes Generated files - skip generated files, except for vendored files

!vendor/**/zz_generated.*

# editor and IDE paraphernalia
.idea
*.swp
*.swo
*~
"""
    return File(".gitignore", content, IfExists.SKIP)


def dockerignore(ctx: Context) -> File:
    content = """# More info: https://docs.docker.com/engine/reference/builder/#dockerignore-file
# Ignore all files which are not go type
!**/*.go
!**/*.mod
!**/*.sum
"""
    return File(".dockerignore", content, IfExists.SKIP)


def config_default_kustomization(ctx: Context) -> File:
    name = ctx.project_name
    content = f"""# Adds namespace to all resources.
namespace: {name}-system

# Value of this field is prepended to the
# names of all resources, e.g. a deployment named
# "wordpress" becomes "alices-wordpress".
# Note that it should also match with the prefix (text before '-') of the namespace
# field above.
namePrefix: {name}-

# Labels to add to all resources and selectors.
#commonLabels:
#  someName: someValue

bases:
- ../crd
- ../rbac
- ../manager
# [WEBHOOK] To enable webhook, uncomment all the sections with [WEBHOOK] prefix including the one in
# crd/kustomization.yaml
#- ../webhook
# [CERTMANAGER] To enable cert-manager, uncomment all sections with 'CERTMANAGER'. 'WEBHOOK' components are required.
#- ../certmanager
# [PROMETHEUS] To enable prometheus monitor, uncomment all sections with 'PROMETHEUS'.
#- ../prometheus

patchesStrategicMerge:
# Protect the /metrics endpoint by putting it behind auth.
# If you want your controller-manager to expose the /metrics
# endpoint w/o any authn/z, please comment the following line.
- manager_auth_proxy_patch.yaml

# Mount the controller config file for loading manager configurations
# through a ComponentConfig type
#- manager_config_patch.yaml
"""
    return File(
        os.path.join("config", "default", "kustomization.yaml"),
        content,
        IfExists.SKIP,
    )


def config_default_auth_proxy_patch(ctx: Context) -> File:
    content = """# This patch inject a sidecar container which is a HTTP proxy for the
# controller manager, it performs RBAC authorization against the Kubernetes API using SubjectAccessReviews.
apiVersion: apps/v1
kind: Deployment
metadata:
  name: controller-manager
  namespace: system
spec:
  template:
    spec:
      containers:
      - name: kube-rbac-proxy
        image: gcr.io/kubebuilder/kube-rbac-proxy:v0.8.0
        args:
        - "--secure-listen-address=0.0.0.0:8443"
        - "--upstream=http://127.0.0.1:8080/"
        - "--logtostderr=true"
        - "--v=10"
        ports:
        - containerPort: 8443
          protocol: TCP
          name: https
      - name: manager
        args:
        - "--health-probe-bind-address=:8081"
        - "--metrics-bind-address=127.0.0.1:8080"
        - "--leader-elect"
"""
    return File(
        os.path.join("config", "default", "manager_auth_proxy_patch.yaml"),
        content,
        IfExists.SKIP,
    )


def config_default_manager_config_patch(ctx: Context) -> File:
    content = """apiVersion: apps/v1
kind: Deployment
metadata:
  name: controller-manager
  namespace: system
spec:
  template:
    spec:
      containers:
      - name: manager
        args:
        - "--config=controller_manager_config.yaml"
        volumeMounts:
        - name: manager-config
          mountPath: /controller_manager_config.yaml
          subPath: controller_manager_config.yaml
      volumes:
      - name: manager-config
        configMap:
          name: manager-config
"""
    return File(
        os.path.join("config", "default", "manager_config_patch.yaml"),
        content,
        IfExists.SKIP,
    )


def config_manager_kustomization(ctx: Context) -> File:
    content = """resources:
- manager.yaml

generatorOptions:
  disableNameSuffixHash: true

configMapGenerator:
- name: manager-config
  files:
  - controller_manager_config.yaml

apiVersion: kustomize.config.k8s.io/v1beta1
kind: Kustomization
images:
- name: controller
  newName: controller
  newTag: latest
"""
    return File(
        os.path.join("config", "manager", "kustomization.yaml"),
        content,
        IfExists.SKIP,
    )


def config_manager_manager(ctx: Context) -> File:
    content = """apiVersion: v1
kind: Namespace
metadata:
  labels:
    control-plane: controller-manager
  name: system
---
apiVersion: apps/v1
kind: Deployment
metadata:
  name: controller-manager
  namespace: system
  labels:
    control-plane: controller-manager
spec:
  selector:
    matchLabels:
      control-plane: controller-manager
  replicas: 1
  template:
    metadata:
      labels:
        control-plane: controller-manager
    spec:
      securityContext:
        runAsNonRoot: true
      containers:
      - command:
        - /manager
        args:
        - --leader-elect
        image: controller:latest
        name: manager
        securityContext:
          allowPrivilegeEscalation: false
        livenessProbe:
          httpGet:
            path: /healthz
            port: 8081
          initialDelaySeconds: 15
          periodSeconds: 20
        readinessProbe:
          httpGet:
            path: /readyz
            port: 8081
          initialDelaySeconds: 5
          periodSeconds: 10
        resources:
          limits:
            cpu: 200m
            memory: 100Mi
          requests:
            cpu: 100m
            memory: 20Mi
      serviceAccountName: controller-manager
      terminationGracePeriodSeconds: 10
"""
    return File(
        os.path.join("config", "manager", "manager.yaml"),
        content,
        IfExists.SKIP,
    )


def config_manager_controller_manager_config(ctx: Context) -> File:
    content = f"""apiVersion: controller-runtime.sigs.k8s.io/v1alpha1
kind: ControllerManagerConfig
health:
  healthProbeBindAddress: :8081
metrics:
  bindAddress: 127.0.0.1:8080
webhook:
  port: 9443
leaderElection:
  leaderElect: true
  resourceName: {ctx.project_name}-leader-election
"""
    return File(
        os.path.join(
            "config", "manager", "controller_manager_config.yaml"
        ),
        content,
        IfExists.SKIP,
    )


def config_prometheus(ctx: Context) -> list[File]:
    kustomization = File(
        os.path.join("config", "prometheus", "kustomization.yaml"),
        "resources:\n- monitor.yaml\n",
        IfExists.SKIP,
    )
    monitor = File(
        os.path.join("config", "prometheus", "monitor.yaml"),
        """
# Prometheus Monitor Service (Metrics)
apiVersion: monitoring.coreos.com/v1
kind: ServiceMonitor
metadata:
  labels:
    control-plane: controller-manager
  name: controller-manager-metrics-monitor
  namespace: system
spec:
  endpoints:
    - path: /metrics
      port: https
      scheme: https
      bearerTokenFile: /var/run/secrets/kubernetes.io/serviceaccount/token
      tlsConfig:
        insecureSkipVerify: true
  selector:
    matchLabels:
      control-plane: controller-manager
""",
        IfExists.SKIP,
    )
    return [kustomization, monitor]


def config_rbac(ctx: Context) -> list[File]:
    files = []

    files.append(
        File(
            os.path.join("config", "rbac", "kustomization.yaml"),
            """resources:
# All RBAC will be applied under this service account in
# the deployment namespace. You may comment out this resource
# if your manager will use a service account that exists at
# runtime. Be sure to update RoleBinding and ClusterRoleBinding
# subjects if changing service account names.
- service_account.yaml
- role.yaml
- role_binding.yaml
- leader_election_role.yaml
- leader_election_role_binding.yaml
# Comment the following 4 lines if you want to disable
# the auth proxy (https://github.com/brancz/kube-rbac-proxy)
# which protects your /metrics endpoint.
- auth_proxy_service.yaml
- auth_proxy_role.yaml
- auth_proxy_role_binding.yaml
- auth_proxy_client_clusterrole.yaml
""",
            IfExists.SKIP,
        )
    )

    files.append(
        File(
            os.path.join("config", "rbac", "service_account.yaml"),
            """apiVersion: v1
kind: ServiceAccount
metadata:
  name: controller-manager
  namespace: system
""",
            IfExists.SKIP,
        )
    )

    files.append(
        File(
            os.path.join("config", "rbac", "role_binding.yaml"),
            """apiVersion: rbac.authorization.k8s.io/v1
kind: ClusterRoleBinding
metadata:
  name: manager-rolebinding
roleRef:
  apiGroup: rbac.authorization.k8s.io
  kind: ClusterRole
  name: manager-role
subjects:
- kind: ServiceAccount
  name: controller-manager
  namespace: system
""",
            IfExists.SKIP,
        )
    )

    files.append(
        File(
            os.path.join("config", "rbac", "leader_election_role.yaml"),
            """# permissions to do leader election.
apiVersion: rbac.authorization.k8s.io/v1
kind: Role
metadata:
  name: leader-election-role
rules:
- apiGroups:
  - ""
  resources:
  - configmaps
  verbs:
  - get
  - list
  - watch
  - create
  - update
  - patch
  - delete
- apiGroups:
  - coordination.k8s.io
  resources:
  - leases
  verbs:
  - get
  - list
  - watch
  - create
  - update
  - patch
  - delete
- apiGroups:
  - ""
  resources:
  - events
  verbs:
  - create
  - patch
""",
            IfExists.SKIP,
        )
    )

    files.append(
        File(
            os.path.join(
                "config", "rbac", "leader_election_role_binding.yaml"
            ),
            """apiVersion: rbac.authorization.k8s.io/v1
kind: RoleBinding
metadata:
  name: leader-election-rolebinding
roleRef:
  apiGroup: rbac.authorization.k8s.io
  kind: Role
  name: leader-election-role
subjects:
- kind: ServiceAccount
  name: controller-manager
  namespace: system
""",
            IfExists.SKIP,
        )
    )

    files.append(
        File(
            os.path.join("config", "rbac", "auth_proxy_service.yaml"),
            """apiVersion: v1
kind: Service
metadata:
  labels:
    control-plane: controller-manager
  name: controller-manager-metrics-service
  namespace: system
spec:
  ports:
  - name: https
    port: 8443
    protocol: TCP
    targetPort: https
  selector:
    control-plane: controller-manager
""",
            IfExists.SKIP,
        )
    )

    files.append(
        File(
            os.path.join("config", "rbac", "auth_proxy_role.yaml"),
            """apiVersion: rbac.authorization.k8s.io/v1
kind: ClusterRole
metadata:
  name: proxy-role
rules:
- apiGroups:
  - authentication.k8s.io
  resources:
  - tokenreviews
  verbs:
  - create
- apiGroups:
  - authorization.k8s.io
  resources:
  - subjectaccessreviews
  verbs:
  - create
""",
            IfExists.SKIP,
        )
    )

    files.append(
        File(
            os.path.join("config", "rbac", "auth_proxy_role_binding.yaml"),
            """apiVersion: rbac.authorization.k8s.io/v1
kind: ClusterRoleBinding
metadata:
  name: proxy-rolebinding
roleRef:
  apiGroup: rbac.authorization.k8s.io
  kind: ClusterRole
  name: proxy-role
subjects:
- kind: ServiceAccount
  name: controller-manager
  namespace: system
""",
            IfExists.SKIP,
        )
    )

    files.append(
        File(
            os.path.join(
                "config", "rbac", "auth_proxy_client_clusterrole.yaml"
            ),
            """apiVersion: rbac.authorization.k8s.io/v1
kind: ClusterRole
metadata:
  name: metrics-reader
rules:
- nonResourceURLs:
  - "/metrics"
  verbs:
  - get
""",
            IfExists.SKIP,
        )
    )

    # placeholder aggregate role; controller-gen regenerates this from the
    # +kubebuilder:rbac markers via `make manifests`
    files.append(
        File(
            os.path.join("config", "rbac", "role.yaml"),
            """apiVersion: rbac.authorization.k8s.io/v1
kind: ClusterRole
metadata:
  creationTimestamp: null
  name: manager-role
rules:
- apiGroups:
  - '*'
  resources:
  - '*'
  verbs:
  - '*'
""",
            IfExists.SKIP,
        )
    )

    return files


def config_crd_kustomizeconfig(ctx: Context) -> File:
    content = """# This file is for teaching kustomize how to substitute name and namespace reference in CRD
nameReference:
- kind: Service
  version: v1
  fieldSpecs:
  - kind: CustomResourceDefinition
    version: v1
    group: apiextensions.k8s.io
    path: spec/conversion/webhook/clientConfig/service/name

namespace:
- kind: CustomResourceDefinition
  version: v1
  group: apiextensions.k8s.io
  path: spec/conversion/webhook/clientConfig/service/namespace
  create: false

varReference:
- path: metadata/annotations
"""
    return File(
        os.path.join("config", "crd", "kustomizeconfig.yaml"),
        content,
        IfExists.SKIP,
    )


def rbac_editor_role(ctx: Context) -> File:
    res = ctx.resource
    content = f"""# permissions for end users to edit {res.plural}.
apiVersion: rbac.authorization.k8s.io/v1
kind: ClusterRole
metadata:
  name: {res.kind.lower()}-editor-role
rules:
- apiGroups:
  - {res.qualified_group}
  resources:
  - {res.plural}
  verbs:
  - create
  - delete
  - get
  - list
  - patch
  - update
  - watch
- apiGroups:
  - {res.qualified_group}
  resources:
  - {res.plural}/status
  verbs:
  - get
"""
    return File(
        os.path.join(
            "config", "rbac", f"{res.kind.lower()}_editor_role.yaml"
        ),
        content,
        IfExists.SKIP,
    )


def rbac_viewer_role(ctx: Context) -> File:
    res = ctx.resource
    content = f"""# permissions for end users to view {res.plural}.
apiVersion: rbac.authorization.k8s.io/v1
kind: ClusterRole
metadata:
  name: {res.kind.lower()}-viewer-role
rules:
- apiGroups:
  - {res.qualified_group}
  resources:
  - {res.plural}
  verbs:
  - get
  - list
  - watch
- apiGroups:
  - {res.qualified_group}
  resources:
  - {res.plural}/status
  verbs:
  - get
"""
    return File(
        os.path.join(
            "config", "rbac", f"{res.kind.lower()}_viewer_role.yaml"
        ),
        content,
        IfExists.SKIP,
    )


def init_files(ctx: Context) -> list[File]:
    """Everything the base plugins lay down at `init` time."""
    return [
        gitignore(ctx),
        dockerignore(ctx),
        config_default_kustomization(ctx),
        config_default_auth_proxy_patch(ctx),
        config_default_manager_config_patch(ctx),
        config_manager_kustomization(ctx),
        config_manager_manager(ctx),
        config_manager_controller_manager_config(ctx),
        *config_prometheus(ctx),
        *config_rbac(ctx),
        config_crd_kustomizeconfig(ctx),
    ]
