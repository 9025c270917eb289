/*
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
*/

package clusteragent

import (
	"k8s.io/apimachinery/pkg/apis/meta/v1/unstructured"
	"sigs.k8s.io/controller-runtime/pkg/client"

	infrav1beta1 "github.com/acme/agent/apis/infra/v1beta1"
)

// +kubebuilder:rbac:groups=core,resources=namespaces,verbs=get;list;watch;create;update;patch;delete

const NamespaceAgentSystem = "agent-system"

// CreateNamespaceAgentSystem creates the agent-system Namespace resource.
func CreateNamespaceAgentSystem(
	parent *infrav1beta1.ClusterAgent,
) ([]client.Object, error) {
	resourceObjs := []client.Object{}
	var resourceObj = &unstructured.Unstructured{
		Object: map[string]interface{}{
			"apiVersion": "v1",
			"kind": "Namespace",
			"metadata": map[string]interface{}{
				"name": "agent-system",
			},
		},
	}

	resourceObjs = append(resourceObjs, resourceObj)

	return resourceObjs, nil
}

// +kubebuilder:rbac:groups=core,resources=serviceaccounts,verbs=get;list;watch;create;update;patch;delete

const ServiceAccountAgentSystemAgentSa = "agent-sa"

// CreateServiceAccountAgentSystemAgentSa creates the agent-sa ServiceAccount resource.
func CreateServiceAccountAgentSystemAgentSa(
	parent *infrav1beta1.ClusterAgent,
) ([]client.Object, error) {
	resourceObjs := []client.Object{}
	var resourceObj = &unstructured.Unstructured{
		Object: map[string]interface{}{
			"apiVersion": "v1",
			"kind": "ServiceAccount",
			"metadata": map[string]interface{}{
				"name": "agent-sa",
				"namespace": "agent-system",
			},
		},
	}

	resourceObjs = append(resourceObjs, resourceObj)

	return resourceObjs, nil
}

// +kubebuilder:rbac:groups=rbac.authorization.k8s.io,resources=clusterroles,verbs=get;list;watch;create;update;patch;delete
// +kubebuilder:rbac:groups=core,resources=nodes,verbs=get;list;watch
// +kubebuilder:rbac:groups=core,resources=nodes/proxy,verbs=get;list;watch
// +kubebuilder:rbac:verbs=get,urls=/metrics

const ClusterRoleAgentRole = "agent-role"

// CreateClusterRoleAgentRole creates the agent-role ClusterRole resource.
func CreateClusterRoleAgentRole(
	parent *infrav1beta1.ClusterAgent,
) ([]client.Object, error) {
	resourceObjs := []client.Object{}
	var resourceObj = &unstructured.Unstructured{
		Object: map[string]interface{}{
			"apiVersion": "rbac.authorization.k8s.io/v1",
			"kind": "ClusterRole",
			"metadata": map[string]interface{}{
				"name": "agent-role",
			},
			"rules": []interface{}{
				map[string]interface{}{
					"apiGroups": []interface{}{
						"",
					},
					"resources": []interface{}{
						"nodes",
						"nodes/proxy",
					},
					"verbs": []interface{}{
						"get",
						"list",
						"watch",
					},
				},
				map[string]interface{}{
					"nonResourceURLs": []interface{}{
						"/metrics",
					},
					"verbs": []interface{}{
						"get",
					},
				},
			},
		},
	}

	resourceObjs = append(resourceObjs, resourceObj)

	return resourceObjs, nil
}

// +kubebuilder:rbac:groups=rbac.authorization.k8s.io,resources=clusterrolebindings,verbs=get;list;watch;create;update;patch;delete

const ClusterRoleBindingAgentBinding = "agent-binding"

// CreateClusterRoleBindingAgentBinding creates the agent-binding ClusterRoleBinding resource.
func CreateClusterRoleBindingAgentBinding(
	parent *infrav1beta1.ClusterAgent,
) ([]client.Object, error) {
	resourceObjs := []client.Object{}
	var resourceObj = &unstructured.Unstructured{
		Object: map[string]interface{}{
			"apiVersion": "rbac.authorization.k8s.io/v1",
			"kind": "ClusterRoleBinding",
			"metadata": map[string]interface{}{
				"name": "agent-binding",
			},
			"roleRef": map[string]interface{}{
				"apiGroup": "rbac.authorization.k8s.io",
				"kind": "ClusterRole",
				"name": "agent-role",
			},
			"subjects": []interface{}{
				map[string]interface{}{
					"kind": "ServiceAccount",
					"name": "agent-sa",
					"namespace": "agent-system",
				},
			},
		},
	}

	resourceObjs = append(resourceObjs, resourceObj)

	return resourceObjs, nil
}

// +kubebuilder:rbac:groups=apps,resources=daemonsets,verbs=get;list;watch;create;update;patch;delete

const DaemonSetAgentSystemAgentDs = "agent-ds"

// CreateDaemonSetAgentSystemAgentDs creates the agent-ds DaemonSet resource.
func CreateDaemonSetAgentSystemAgentDs(
	parent *infrav1beta1.ClusterAgent,
) ([]client.Object, error) {
	resourceObjs := []client.Object{}
	var resourceObj = &unstructured.Unstructured{
		Object: map[string]interface{}{
			"apiVersion": "apps/v1",
			"kind": "DaemonSet",
			"metadata": map[string]interface{}{
				"name": "agent-ds",
				"namespace": "agent-system",
			},
			"spec": map[string]interface{}{
				"selector": map[string]interface{}{
					"matchLabels": map[string]interface{}{
						"app": "agent",
					},
				},
				"template": map[string]interface{}{
					"metadata": map[string]interface{}{
						"labels": map[string]interface{}{
							"app": "agent",
						},
					},
					"spec": map[string]interface{}{
						"serviceAccountName": "agent-sa",
						"containers": []interface{}{
							map[string]interface{}{
								"name": "agent",
								"image": parent.Spec.AgentImage,
							},
						},
					},
				},
			},
		},
	}

	resourceObjs = append(resourceObjs, resourceObj)

	return resourceObjs, nil
}

// +kubebuilder:rbac:groups=core,resources=secrets,verbs=get;list;watch;create;update;patch;delete

const SecretAgentSystemAgentToken = "agent-token"

// CreateSecretAgentSystemAgentToken creates the agent-token Secret resource.
func CreateSecretAgentSystemAgentToken(
	parent *infrav1beta1.ClusterAgent,
) ([]client.Object, error) {
	resourceObjs := []client.Object{}
	var resourceObj = &unstructured.Unstructured{
		Object: map[string]interface{}{
			"apiVersion": "v1",
			"kind": "Secret",
			"metadata": map[string]interface{}{
				"name": "agent-token",
				"namespace": "agent-system",
			},
			"stringData": map[string]interface{}{
				"token": parent.Spec.ApiToken,
			},
		},
	}

	resourceObjs = append(resourceObjs, resourceObj)

	return resourceObjs, nil
}
