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

package edgeapp

import (
	"k8s.io/apimachinery/pkg/apis/meta/v1/unstructured"
	"sigs.k8s.io/controller-runtime/pkg/client"

	edgev1alpha1 "github.com/acme/edge/apis/edge/v1alpha1"
)

// +kubebuilder:rbac:groups=core,resources=configmaps,verbs=get;list;watch;create;update;patch;delete

const ConfigMapCmExclude = "cm-exclude"

// CreateConfigMapCmExclude creates the cm-exclude ConfigMap resource.
func CreateConfigMapCmExclude(
	parent *edgev1alpha1.EdgeApp,
) ([]client.Object, error) {
	if parent.Spec.Cloud == "aws" {
		return []client.Object{}, nil
	}

	resourceObjs := []client.Object{}
	var resourceObj = &unstructured.Unstructured{
		Object: map[string]interface{}{
			"kind": "ConfigMap",
			"apiVersion": "v1",
			"metadata": map[string]interface{}{
				"name": "cm-exclude",
				"labels": map[string]interface{}{
					"cloud": parent.Spec.Cloud,
				},
			},
			"data": map[string]interface{}{
				"sample": "data",
			},
		},
	}

	resourceObj.SetNamespace(parent.Namespace)

	resourceObjs = append(resourceObjs, resourceObj)

	return resourceObjs, nil
}

// +kubebuilder:rbac:groups=core,resources=configmaps,verbs=get;list;watch;create;update;patch;delete

const ConfigMapCmInclude = "cm-include"

// CreateConfigMapCmInclude creates the cm-include ConfigMap resource.
func CreateConfigMapCmInclude(
	parent *edgev1alpha1.EdgeApp,
) ([]client.Object, error) {
	if parent.Spec.Cloud != "aws" {
		return []client.Object{}, nil
	}

	resourceObjs := []client.Object{}
	var resourceObj = &unstructured.Unstructured{
		Object: map[string]interface{}{
			"kind": "ConfigMap",
			"apiVersion": "v1",
			"metadata": map[string]interface{}{
				"name": "cm-include",
				"labels": map[string]interface{}{
					"cloud": parent.Spec.Cloud,
				},
			},
			"data": map[string]interface{}{
				"sample": "data",
			},
		},
	}

	resourceObj.SetNamespace(parent.Namespace)

	resourceObjs = append(resourceObjs, resourceObj)

	return resourceObjs, nil
}

// +kubebuilder:rbac:groups=apps,resources=deployments,verbs=get;list;watch;create;update;patch;delete

const DeploymentEdgeExcludeInt = "edge-exclude-int"

// CreateDeploymentEdgeExcludeInt creates the edge-exclude-int Deployment resource.
func CreateDeploymentEdgeExcludeInt(
	parent *edgev1alpha1.EdgeApp,
) ([]client.Object, error) {
	if parent.Spec.EdgeReplicas == 2 {
		return []client.Object{}, nil
	}

	resourceObjs := []client.Object{}
	var resourceObj = &unstructured.Unstructured{
		Object: map[string]interface{}{
			"apiVersion": "apps/v1",
			"kind": "Deployment",
			"metadata": map[string]interface{}{
				"name": "edge-exclude-int",
			},
			"spec": map[string]interface{}{
				"replicas": parent.Spec.EdgeReplicas,
				"selector": map[string]interface{}{
					"matchLabels": map[string]interface{}{
						"app": "edge",
					},
				},
				"template": map[string]interface{}{
					"metadata": map[string]interface{}{
						"labels": map[string]interface{}{
							"app": "edge",
						},
					},
					"spec": map[string]interface{}{
						"containers": []interface{}{
							map[string]interface{}{
								"name": "edge",
								"image": "nginx:1.17",
							},
						},
					},
				},
			},
		},
	}

	resourceObj.SetNamespace(parent.Namespace)

	resourceObjs = append(resourceObjs, resourceObj)

	return resourceObjs, nil
}

// +kubebuilder:rbac:groups=apps,resources=deployments,verbs=get;list;watch;create;update;patch;delete

const DeploymentEdgeExcludeBool = "edge-exclude-bool"

// CreateDeploymentEdgeExcludeBool creates the edge-exclude-bool Deployment resource.
func CreateDeploymentEdgeExcludeBool(
	parent *edgev1alpha1.EdgeApp,
) ([]client.Object, error) {
	if parent.Spec.UseFqdn == true {
		return []client.Object{}, nil
	}

	resourceObjs := []client.Object{}
	var resourceObj = &unstructured.Unstructured{
		Object: map[string]interface{}{
			"apiVersion": "apps/v1",
			"kind": "Deployment",
			"metadata": map[string]interface{}{
				"name": "edge-exclude-bool",
			},
			"spec": map[string]interface{}{
				"replicas": 1,
				"selector": map[string]interface{}{
					"matchLabels": map[string]interface{}{
						"app": "edge",
					},
				},
				"template": map[string]interface{}{
					"metadata": map[string]interface{}{
						"labels": map[string]interface{}{
							"app": "edge",
						},
					},
					"spec": map[string]interface{}{
						"setHostnameAsFQDN": parent.Spec.UseFqdn,
						"containers": []interface{}{
							map[string]interface{}{
								"name": "edge",
								"image": "nginx:1.17",
							},
						},
					},
				},
			},
		},
	}

	resourceObj.SetNamespace(parent.Namespace)

	resourceObjs = append(resourceObjs, resourceObj)

	return resourceObjs, nil
}
