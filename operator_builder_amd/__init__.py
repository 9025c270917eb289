"""operator_builder_amd — a from-scratch re-implementation of the
operator-builder capability surface (reference: vmware-tanzu-labs/operator-builder).

The reference is a pure-Go, CPU-only CLI code generator:
it takes a YAML workload config plus marker-annotated Kubernetes manifests and
generates the full Go source tree of a Kubernetes operator.  There is no GPU,
tensor, or distributed-compute surface to map onto MI355X hardware (SURVEY.md
§2.7, §5), so this rebuild is a CPU-only, single-process code generator with
the same commands, marker language, and generated-source layout.

Package map (reference parity):
  utils/     -> internal/utils           (name casing, ** glob, conversions)
  markers/   -> internal/markers         (generic lexer/parser/registry/inspector)
  yamlast/   -> gopkg.in/yaml.v3 yaml.Node (comment-preserving YAML round-trip)
  workload/  -> internal/workload/v1     (config, kinds, manifests, markers, rbac)
  codegen/   -> object-code-generator-for-k8s (YAML -> Go object source)
  scaffold/  -> internal/plugins/workload/v1/scaffolds + kubebuilder machinery
  license/   -> internal/license
  cli/       -> cmd/operator-builder + pkg/cli
"""

__version__ = "0.2.0"
