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

package mesh

import (
	"k8s.io/apimachinery/pkg/apis/meta/v1/unstructured"
	"sigs.k8s.io/controller-runtime/pkg/client"

	edgeplatformv1alpha1 "github.com/acme/edgeplatform/apis/edgeplatform/v1alpha1"
	meshv1alpha1 "github.com/acme/edgeplatform/apis/mesh/v1alpha1"
)

// +kubebuilder:rbac:groups=rbac.authorization.k8s.io,resources=clusterroles,verbs=get;list;watch;create;update;patch;delete
// +kubebuilder:rbac:groups=core,resources=pods,verbs=get;list;watch
// +kubebuilder:rbac:groups=core,resources=endpoints,verbs=get;list;watch

const ClusterRoleMeshAgent = "mesh-agent"

// CreateClusterRoleMeshAgent creates the mesh-agent ClusterRole resource.
func CreateClusterRoleMeshAgent(
	parent *meshv1alpha1.ServiceMesh,
	collection *edgeplatformv1alpha1.EdgePlatform,
) ([]client.Object, error) {
	resourceObjs := []client.Object{}
	var resourceObj = &unstructured.Unstructured{
		Object: map[string]interface{}{
			"apiVersion": "rbac.authorization.k8s.io/v1",
			"kind": "ClusterRole",
			"metadata": map[string]interface{}{
				"name": "mesh-agent",
			},
			"rules": []interface{}{
				map[string]interface{}{
					"apiGroups": []interface{}{
						"",
					},
					"resources": []interface{}{
						"pods",
						"endpoints",
					},
					"verbs": []interface{}{
						"get",
						"list",
						"watch",
					},
				},
			},
		},
	}

	resourceObjs = append(resourceObjs, resourceObj)

	return resourceObjs, nil
}

// +kubebuilder:rbac:groups=apps,resources=daemonsets,verbs=get;list;watch;create;update;patch;delete

const DaemonSetEdgeSystemMeshAgent = "mesh-agent"

// CreateDaemonSetEdgeSystemMeshAgent creates the mesh-agent DaemonSet resource.
func CreateDaemonSetEdgeSystemMeshAgent(
	parent *meshv1alpha1.ServiceMesh,
	collection *edgeplatformv1alpha1.EdgePlatform,
) ([]client.Object, error) {
	resourceObjs := []client.Object{}
	var resourceObj = &unstructured.Unstructured{
		Object: map[string]interface{}{
			"apiVersion": "apps/v1",
			"kind": "DaemonSet",
			"metadata": map[string]interface{}{
				"name": "mesh-agent",
				"namespace": "edge-system",
			},
			"spec": map[string]interface{}{
				"selector": map[string]interface{}{
					"matchLabels": map[string]interface{}{
						"app": "mesh-agent",
					},
				},
				"template": map[string]interface{}{
					"metadata": map[string]interface{}{
						"labels": map[string]interface{}{
							"app": "mesh-agent",
						},
					},
					"spec": map[string]interface{}{
						"serviceAccountName": "mesh-agent",
						"containers": []interface{}{
							map[string]interface{}{
								"name": "agent",
								"image": parent.Spec.MeshAgentImage,
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

// +kubebuilder:rbac:groups=apiextensions.k8s.io,resources=customresourcedefinitions,verbs=get;list;watch;create;update;patch;delete

const CustomResourceDefinitionMeshpoliciesMeshExampleCom = "meshpolicies.mesh.example.com"

// CreateCustomResourceDefinitionMeshpoliciesMeshExampleCom creates the meshpolicies.mesh.example.com CustomResourceDefinition resource.
func CreateCustomResourceDefinitionMeshpoliciesMeshExampleCom(
	parent *meshv1alpha1.ServiceMesh,
	collection *edgeplatformv1alpha1.EdgePlatform,
) ([]client.Object, error) {
	resourceObjs := []client.Object{}
	var resourceObj = &unstructured.Unstructured{
		Object: map[string]interface{}{
			"apiVersion": "apiextensions.k8s.io/v1",
			"kind": "CustomResourceDefinition",
			"metadata": map[string]interface{}{
				"name": "meshpolicies.mesh.example.com",
			},
			"spec": map[string]interface{}{
				"group": "mesh.example.com",
				"names": map[string]interface{}{
					"kind": "MeshPolicy",
					"listKind": "MeshPolicyList",
					"plural": "meshpolicies",
					"singular": "meshpolicy",
				},
				"scope": "Namespaced",
				"versions": []interface{}{
					map[string]interface{}{
						"name": "v1",
						"served": true,
						"storage": true,
						"schema": map[string]interface{}{
							"openAPIV3Schema": map[string]interface{}{
								"type": "object",
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
