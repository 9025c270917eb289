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

package webapp

import (
	"k8s.io/apimachinery/pkg/apis/meta/v1/unstructured"
	"sigs.k8s.io/controller-runtime/pkg/client"

	appsv1alpha1 "github.com/acme/platform/apis/apps/v1alpha1"
	platformsv1alpha1 "github.com/acme/platform/apis/platforms/v1alpha1"
)

// +kubebuilder:rbac:groups=apps,resources=deployments,verbs=get;list;watch;create;update;patch;delete

const DeploymentWebappDeploy = "webapp-deploy"

// CreateDeploymentWebappDeploy creates the webapp-deploy Deployment resource.
func CreateDeploymentWebappDeploy(
	parent *appsv1alpha1.WebApp,
	collection *platformsv1alpha1.CloudPlatform,
) ([]client.Object, error) {
	if parent.Spec.DeployWebApp != true {
		return []client.Object{}, nil
	}

	resourceObjs := []client.Object{}
	var resourceObj = &unstructured.Unstructured{
		Object: map[string]interface{}{
			"apiVersion": "apps/v1",
			"kind": "Deployment",
			"metadata": map[string]interface{}{
				"name": "webapp-deploy",
				"labels": map[string]interface{}{
					"environment": collection.Spec.Environment,
				},
			},
			"spec": map[string]interface{}{
				"replicas": parent.Spec.WebAppReplicas,
				"selector": map[string]interface{}{
					"matchLabels": map[string]interface{}{
						"app": "webapp",
					},
				},
				"template": map[string]interface{}{
					"metadata": map[string]interface{}{
						"labels": map[string]interface{}{
							"app": "webapp",
						},
					},
					"spec": map[string]interface{}{
						"containers": []interface{}{
							map[string]interface{}{
								"name": "webapp",
								"image": parent.Spec.WebAppImage,
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

// +kubebuilder:rbac:groups=core,resources=configmaps,verbs=get;list;watch;create;update;patch;delete

const ConfigMapWebappSettings = "webapp-settings"

// CreateConfigMapWebappSettings creates the webapp-settings ConfigMap resource.
func CreateConfigMapWebappSettings(
	parent *appsv1alpha1.WebApp,
	collection *platformsv1alpha1.CloudPlatform,
) ([]client.Object, error) {
	resourceObjs := []client.Object{}
	var resourceObj = &unstructured.Unstructured{
		Object: map[string]interface{}{
			"apiVersion": "v1",
			"kind": "ConfigMap",
			"metadata": map[string]interface{}{
				"name": "webapp-settings",
			},
			"data": map[string]interface{}{
				"deploy": parent.Spec.DeployWebApp,
			},
		},
	}

	resourceObj.SetNamespace(parent.Namespace)

	resourceObjs = append(resourceObjs, resourceObj)

	return resourceObjs, nil
}
